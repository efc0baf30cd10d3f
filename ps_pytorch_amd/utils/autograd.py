"""Direct backward driving (reference parity:
src/distributed_functions/distributed_backward.py).

The reference vendored torch-0.3's `torch.autograd.backward` to expose
`run_backward(variables, grad_variables)` so a trainer can start backprop
from non-scalar outputs with explicit seed gradients (used by
single_machine.py's LeNetLearner:64-69 pattern: compute the loss on a
DETACHED copy of the logits, then backward the live graph from
`logits_copy.grad`). Modern torch does this natively; these helpers keep the
reference's call shape and add the detached-loss idiom as one function —
the same trick the split models use to overlap loss computation with the
gradient push (resnet_split.py:188-251)."""
from __future__ import annotations

from typing import Optional, Sequence, Union

import torch

Tensors = Union[torch.Tensor, Sequence[torch.Tensor]]


def run_backward(tensors: Tensors, grad_tensors: Optional[Tensors] = None,
                 retain_graph: bool = False, create_graph: bool = False) -> None:
    """ref distributed_backward.py:38-90 `backward` — explicit-seed backprop."""
    torch.autograd.backward(tensors, grad_tensors, retain_graph=retain_graph,
                            create_graph=create_graph)


def loss_backward_via_logits(network_out: torch.Tensor, loss_fn,
                             target: torch.Tensor) -> torch.Tensor:
    """Compute loss on a detached logits copy, then backward the live graph
    from the copy's gradient (ref single_machine.py:64-69). Returns the
    detached loss."""
    logits = network_out.detach().requires_grad_(True)
    loss = loss_fn(logits, target)
    loss.backward()
    run_backward(network_out, logits.grad)
    return loss.detach()
