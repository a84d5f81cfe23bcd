"""Flat-tensor cross-rank agent transfer (parallel/flat_transfer.py)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from agilerl_amd.parallel.flat_transfer import merge_tensor_tree, split_tensor_tree


class TestSplitMerge:
    def test_roundtrip_nested(self):
        tree = {
            "a": torch.randn(3, 4),
            "b": {"c": torch.arange(5), "d": [torch.ones(2), "text", 7]},
            "e": (torch.zeros(1, dtype=torch.bool), None),
            "f": 3.14,
        }
        skeleton, tensors = split_tensor_tree(tree)
        assert len(tensors) == 4
        # skeleton contains no tensors
        flat_repr = repr(skeleton)
        assert "tensor" not in flat_repr.lower() or "__flatxfer" in flat_repr
        out = merge_tensor_tree(skeleton, tensors)
        torch.testing.assert_close(out["a"], tree["a"])
        torch.testing.assert_close(out["b"]["c"], tree["b"]["c"])
        torch.testing.assert_close(out["b"]["d"][0], tree["b"]["d"][0])
        assert out["b"]["d"][1] == "text" and out["b"]["d"][2] == 7
        assert out["e"][0].dtype == torch.bool
        assert out["e"][1] is None
        assert out["f"] == 3.14

    def test_dtype_preserved(self):
        tree = {
            "f32": torch.randn(4),
            "f64": torch.randn(4, dtype=torch.float64),
            "i64": torch.tensor([2**40 + 3]),
            "u8": torch.tensor([255], dtype=torch.uint8),
        }
        skeleton, tensors = split_tensor_tree(tree)
        out = merge_tensor_tree(skeleton, tensors)
        for k in tree:
            assert out[k].dtype == tree[k].dtype
            torch.testing.assert_close(out[k], tree[k])


def _bcast_worker(rank, world_size, port, results):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.algorithms.dqn import DQN
    from agilerl_amd.parallel import DistributedState
    from agilerl_amd.parallel.flat_transfer import broadcast_checkpoint
    from agilerl_amd.spaces import Box, Discrete

    DistributedState.reset()
    state = DistributedState.get()

    if rank == 1:  # src is rank 1, NOT 0 — exercises non-main ownership
        torch.manual_seed(42)
        agent = DQN(Box(-1, 1, (4,)), Discrete(2))
        # give the optimizer real moments (int64 step + fp32 exp_avg)
        batch = {
            "obs": torch.randn(8, 4), "action": torch.randint(0, 2, (8, 1)),
            "reward": torch.randn(8, 1), "next_obs": torch.randn(8, 4),
            "done": torch.zeros(8, 1),
        }
        agent.learn(batch)
        ckpt = agent.get_checkpoint_dict()
    else:
        ckpt = None
    out = broadcast_checkpoint(ckpt, src=1, rank=rank,
                               device=state.device, backend=state.backend)
    # every rank reconstructs the identical agent
    agent2 = DQN(Box(-1, 1, (4,)), Discrete(2))
    agent2._apply_checkpoint(out)
    x = torch.randn(5, 4, generator=torch.Generator().manual_seed(9))
    results[rank] = agent2.actor(x).detach().numpy()
    torch.distributed.barrier()
    torch.distributed.destroy_process_group()


def test_broadcast_checkpoint_gloo():
    port = _free_port()
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_bcast_worker, args=(r, 2, port, results))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
            assert p.exitcode == 0
        np.testing.assert_allclose(results[0], results[1], rtol=0, atol=0)


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.gpu
class TestNcclSingleRank:
    """Exercise every collective our paths use under the real RCCL backend
    at world_size=1 — validates RCCL init + the nccl-specific tensor/device
    handling before the driver's first 8-GPU attempt."""

    def test_collectives_and_checkpoint_broadcast(self):
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", str(_free_port()))
        assert torch.cuda.is_available()
        if not dist.is_initialized():
            dist.init_process_group("nccl", rank=0, world_size=1)
        try:
            dev = "cuda:0"
            t = torch.ones(4, device=dev)
            dist.all_reduce(t)
            torch.testing.assert_close(t.cpu(), torch.ones(4))
            dist.broadcast(t, src=0)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            obj = [{"plan": [0, 1, 2]}]
            dist.broadcast_object_list(obj, src=0)
            assert obj[0]["plan"] == [0, 1, 2]
            gather = [torch.zeros(4, device=dev)]
            dist.all_gather(gather, t)
            dist.barrier(device_ids=[0])

            # full checkpoint broadcast through the flat-transfer path
            from agilerl_amd.algorithms.dqn import DQN
            from agilerl_amd.parallel.flat_transfer import broadcast_checkpoint
            from agilerl_amd.spaces import Box, Discrete

            agent = DQN(Box(-1, 1, (4,)), Discrete(2), device=dev)
            out = broadcast_checkpoint(
                agent.get_checkpoint_dict(), src=0, rank=0,
                device=dev, backend="nccl",
            )
            agent2 = DQN(Box(-1, 1, (4,)), Discrete(2), device=dev)
            agent2._apply_checkpoint(out)
            x = torch.randn(5, 4, device=dev)
            torch.testing.assert_close(agent2.actor(x), agent.actor(x))
        finally:
            dist.destroy_process_group()
