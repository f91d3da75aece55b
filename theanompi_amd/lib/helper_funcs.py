"""Checkpoint I/O and small glue helpers.

Rebuild of the reference's ``theanompi/lib/helper_funcs.py`` [pub]
(SURVEY.md §2.3, §5.4): weights and momentum buffers are saved as one
``.npy`` file per parameter, indexed by parameter position and epoch —
``W_<i>_<epoch>.npy`` / ``V_<i>_<epoch>.npy`` — so resume restores true
SGD state (weights + momenta), and checkpoints written by one rule load
under any other.  Tensors are saved as float32 numpy arrays in the
parameter's logical (reference) layout; layout permutation to/from the
kernels' native layouts is the layer's job (see models/layers.py).

The reference's ``bufint``/``dtype_to_mpi`` CUDA-aware-MPI glue has no
equivalent here: RCCL operates on torch tensors directly.
"""

import os

import numpy as np
import torch


def save_weights(params, directory, epoch, prefix="W"):
    """params: list of torch tensors (any device/dtype). One .npy each."""
    os.makedirs(directory, exist_ok=True)
    for i, p in enumerate(params):
        arr = p.detach().to(torch.float32).cpu().numpy()
        np.save(os.path.join(directory, f"{prefix}_{i}_{epoch}.npy"), arr)


def load_weights(params, directory, epoch, prefix="W"):
    """Load .npy files back into the given tensors (in-place)."""
    for i, p in enumerate(params):
        path = os.path.join(directory, f"{prefix}_{i}_{epoch}.npy")
        arr = np.load(path)
        src = torch.from_numpy(arr)
        if tuple(src.shape) != tuple(p.shape):
            raise ValueError(
                f"checkpoint param {i} shape {tuple(src.shape)} != "
                f"model param shape {tuple(p.shape)}")
        with torch.no_grad():
            p.copy_(src.to(p.dtype))


def save_momentums(vels, directory, epoch):
    save_weights(vels, directory, epoch, prefix="V")


def load_momentums(vels, directory, epoch):
    load_weights(vels, directory, epoch, prefix="V")


def checkpoint_exists(directory, epoch, n_params, prefix="W"):
    return all(
        os.path.exists(os.path.join(directory, f"{prefix}_{i}_{epoch}.npy"))
        for i in range(n_params))
