"""2-GPU BSP CIFAR-10 quick-start — the reference README's example
session script, unchanged API."""
from theanompi_amd import BSP

if __name__ == "__main__":
    rule = BSP()
    rule.init(devices=['cuda0', 'cuda1'],
              modelfile='theanompi_amd.models.cifar10',
              modelclass='Cifar10_model',
              epochs=5, sync_type='cdd', exchange_strategy='rccl_bf16')
    rule.wait()
