"""AlexNet GOSGD gossip training on 4 GPUs."""
from theanompi_amd import GOSGD

if __name__ == "__main__":
    rule = GOSGD()
    rule.init(devices=['cuda0', 'cuda1', 'cuda2', 'cuda3'],
              modelfile='theanompi_amd.models.alex_net',
              modelclass='AlexNet',
              epochs=10, p_push=0.05)
    rule.wait()
