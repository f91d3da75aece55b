"""theanompi_amd — an MI355X-native data-parallel CNN training framework.

A from-scratch rebuild of the capabilities of uoguelph-mlrg/Theano-MPI
(see SURVEY.md at the repo root) designed MI355X-first:

  * compute path  : PyTorch-ROCm tensors/autograd + hand-written HIP/CDNA4
                    kernels (MFMA implicit-GEMM conv, fused BN/LRN/softmax,
                    fused SGD) — no MIOpen, no Triton, no hipify.
  * communication : RCCL over xGMI through ``torch.distributed`` (backend
                    "nccl" is RCCL on ROCm); ``gloo`` for the CPU plumbing
                    mode and the asynchronous gossip control plane.
  * launch        : one OS process per GPU, spawned by the rule objects
                    below (the environment has no mpirun; rendezvous is a
                    TCP store on 127.0.0.1).

The user-facing session API mirrors the reference
(Theano-MPI ``theanompi/__init__.py`` [pub]):

    from theanompi_amd import BSP
    rule = BSP()
    rule.init(devices=['cuda0', 'cuda1'],
              modelfile='theanompi_amd.models.cifar10',
              modelclass='Cifar10_model')
    rule.wait()
"""

from theanompi_amd.rules import BSP, EASGD, GOSGD  # noqa: F401

__version__ = "0.1.0"
__all__ = ["BSP", "EASGD", "GOSGD"]
