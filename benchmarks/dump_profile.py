"""Dump the layerwise backward profile (solver inputs) to JSON."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from mgwfbp_amd.dl_trainer import DLTrainer  # noqa: E402
from mgwfbp_amd.profiling import benchmark  # noqa: E402


def main():
    dnn = sys.argv[1] if len(sys.argv) > 1 else 'resnet50'
    dataset = sys.argv[2] if len(sys.argv) > 2 else 'imagenet'
    bs = int(sys.argv[3]) if len(sys.argv) > 3 else 128
    out = sys.argv[4] if len(sys.argv) > 4 else 'gpurun_out/profile.json'
    t = DLTrainer(0, 1, dist=False, batch_size=bs, ngpus=1, data_dir='',
                  dataset=dataset, dnn=dnn, lr=0.01, nworkers=1,
                  prefix='dump', synthetic=True)
    seq, times, sizes = benchmark(t, num_warmup=3, num_iters=20)
    with open(out, 'w') as f:
        json.dump({'dnn': dnn, 'batch_size': bs, 'seq_layernames': seq,
                   'layerwise_times': times, 'sizes': sizes}, f)
    print('dumped %d layers, total backward %.4fs' % (len(seq),
                                                      sum(times)))


if __name__ == '__main__':
    main()
