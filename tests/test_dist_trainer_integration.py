"""End-to-end mgwfbp() entry integration: 2 processes, gloo, a few
training iterations through the REAL distributed entry path (profiling
-> broadcast -> DistributedOptimizer -> train loop)."""
import os

import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['RANK'] = str(rank)
    os.environ['MGX_COMM_BACKEND'] = 'gloo'
    os.environ['MGX_ADAPTIVE_ABC'] = '0'
    import mgwfbp_amd.comm as comm
    from mgwfbp_amd.dist_trainer import mgwfbp
    comm.init()
    trainer = mgwfbp('lenet', 'cifar10', '', world, 0.01, 4, 1,
                     max_epochs=1, max_iters=4)
    flat = torch.cat([p.detach().reshape(-1)
                      for p in trainer.net.parameters()])
    q.put((rank, flat.tolist(), trainer.optimizer.train_iter))
    comm.shutdown()


def test_mgwfbp_entry_two_processes():
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, world, 29621, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    iters = {}
    for _ in range(world):
        rank, flat, ti = q.get()
        res[rank] = torch.tensor(flat)
        iters[rank] = ti
    for p in procs:
        p.join(300)
        assert p.exitcode == 0
    # ranks identical after synchronized training
    assert torch.equal(res[0], res[1])
    assert iters[0] == iters[1] > 0
