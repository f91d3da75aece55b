"""VGG-16 EASGD, 1 server + 7 workers over xGMI (BASELINE config #4)."""
from theanompi_amd import EASGD

if __name__ == "__main__":
    rule = EASGD()
    rule.init(devices=[f'cuda{i}' for i in range(8)],   # cuda0 = server
              modelfile='theanompi_amd.models.vgg16',
              modelclass='VGG16',
              tau=8, alpha=0.5)
    rule.wait()
