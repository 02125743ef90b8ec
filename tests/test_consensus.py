"""Distributed consensus tests: gloo backend, world_size 2, CPU.

Exercises the same code path bench.py uses over RCCL on MI355X
(pack/all-gather/unpack + header-borne all-agreed reduction).
"""

import multiprocessing as mp
import os

import pytest
import torch

from adversarial_spec_amd.parallel.consensus import (
    HDR,
    pack_result,
    unpack_results,
)


class TestPackUnpack:
    def test_roundtrip(self):
        buf = pack_result([5, 6, 7], True, False, 16, torch.device("cpu"))
        res = unpack_results(buf.unsqueeze(0))
        assert res[0].agreed and not res[0].error
        assert res[0].token_ids == [5, 6, 7]

    def test_truncates_to_max(self):
        buf = pack_result(list(range(100)), False, False, 10, torch.device("cpu"))
        res = unpack_results(buf.unsqueeze(0))
        assert res[0].token_ids == list(range(10))

    def test_header_layout(self):
        buf = pack_result([1], True, True, 4, torch.device("cpu"))
        assert buf[0] == 1 and buf[1] == 1 and buf[2] == 1
        assert buf.shape[0] == HDR + 4


def _worker(rank: int, world: int, port: int, q) -> None:
    import torch.distributed as dist

    from adversarial_spec_amd.parallel.consensus import AsyncRoundGather, gather_round

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # rank 0 agrees, rank 1 critiques with tokens
        toks = [10 + rank, 20 + rank, 30 + rank]
        results, all_agreed = gather_round(
            toks, agreed=(rank == 0), error=False, max_tokens=32
        )
        assert len(results) == world
        assert results[0].agreed and not results[1].agreed
        assert results[1].token_ids == [11, 21, 31]
        assert all_agreed is False

        # all agree -> header reduction True
        _, all2 = gather_round([1], agreed=True, error=False, max_tokens=8)
        assert all2 is True

        # errored rank excluded from consensus
        _, all3 = gather_round(
            [2], agreed=(rank == 0), error=(rank == 1), max_tokens=8
        )
        assert all3 is True

        # async API parity on gloo
        g = AsyncRoundGather(max_tokens=8)
        g.launch([rank], agreed=True, error=False)
        res4, all4 = g.wait()
        assert [r.token_ids for r in res4] == [[0], [1]]
        assert all4
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        q.put((rank, f"FAIL: {e}"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_gloo_world2_gather():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29871
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    outcomes = [q.get(timeout=100) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    assert all(o[1] == "ok" for o in outcomes), outcomes
