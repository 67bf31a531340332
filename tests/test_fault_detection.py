"""Fail-fast fault detection: a dead rank must surface as an error on
the surviving ranks within the fabric timeout (the reference's only
fault handling is 240 s ray.get timeouts that crash the run — SURVEY
§5.3; here the process-group timeout plays that role and is
configurable via config["fabric_timeout_s"])."""

import os
import time

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world_size, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from distrl_llm_amd.parallel.fabric import Fabric

    fabric = Fabric(rank, world_size, 1, 1, torch.device("cpu"),
                    timeout_s=8.0)
    if rank == 1:
        return  # dies before participating in the round
    try:
        fabric.broadcast_obj(("round", 1), src=0)  # peers never show up
        fabric.barrier()
        raise AssertionError("collective with a dead peer must not succeed")
    except AssertionError:
        raise
    except Exception:
        raise RuntimeError("peer-timeout-detected")


@pytest.mark.timeout(180)
def test_dead_rank_fails_fast():
    port = 23500 + os.getpid() % 500
    t0 = time.time()
    with pytest.raises(Exception) as ei:
        mp.spawn(_worker, nprocs=2, args=(2, port), join=True)
    elapsed = time.time() - t0
    assert "peer-timeout-detected" in str(ei.value)
    assert elapsed < 120, f"fail-fast took {elapsed:.0f}s"
