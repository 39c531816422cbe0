"""Phase breakdown of the lstman4/an4 training step (why ~6 s/step?)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from mgwfbp_amd.dl_trainer import DLTrainer  # noqa: E402


def bisect(t):
    """Time each stage of the DeepSpeech forward."""
    net = t.net
    (x, input_sizes), (targets, target_sizes) = t.fetch_data()

    def tm(label, fn, *a):
        torch.cuda.synchronize()
        s = time.time()
        out = fn(*a)
        torch.cuda.synchronize()
        print('  %-12s %.3fs' % (label, time.time() - s), flush=True)
        return out

    for it in range(3):
        print(' bisect iter', it)
        h, olens = tm('conv', net.conv, x, input_sizes)
        n, c, f, tt = h.size()
        h = h.view(n, c * f, tt).permute(2, 0, 1).contiguous()
        for i, rnn in enumerate(net.rnns):
            h = tm('rnn%d' % i, rnn, h)
        h = tm('fc', net.fc, h)
        loss = tm('ctc', lambda: t.ctc_loss(h, targets, olens,
                                            target_sizes))
        loss = loss / x.size(0)
        tm('backward', lambda: loss.backward())
        net.zero_grad(set_to_none=False)


def main():
    t = DLTrainer(0, 1, dist=False, batch_size=4, ngpus=1, data_dir='',
                  dataset='an4', dnn='lstman4', lr=2e-4, nworkers=1,
                  prefix='probe', synthetic=True)
    print('model params:', sum(p.numel() for p in t.net.parameters()))
    bisect(t)
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
