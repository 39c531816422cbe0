"""Phase breakdown of the lstman4/an4 training step (why ~6 s/step?)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from mgwfbp_amd.dl_trainer import DLTrainer  # noqa: E402


def main():
    t = DLTrainer(0, 1, dist=False, batch_size=4, ngpus=1, data_dir='',
                  dataset='an4', dnn='lstman4', lr=2e-4, nworkers=1,
                  prefix='probe', synthetic=True)
    print('model params:', sum(p.numel() for p in t.net.parameters()))
    for i in range(8):
        io0, f0, b0 = t.io_time, t.forward_time, t.backward_time
        s = time.time()
        t.zero_grad()
        t.train(1)
        su = time.time()
        t.update_model()
        torch.cuda.synchronize()
        e = time.time()
        print('step %d total %.3fs io %.3f fwd %.3f bwd %.3f upd %.3f'
              % (i, e - s, t.io_time - io0, t.forward_time - f0,
                 t.backward_time - b0, e - su), flush=True)


if __name__ == '__main__':
    main()
