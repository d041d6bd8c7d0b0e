"""RolloutAssembler property tests: uuid assembly, staleness eviction,
done-splice is_fir rewrite (the reference's quirky semantics,
rollout_assembler.py:51-83)."""
import asyncio

import numpy as np
import pytest
import torch

from pdrl_amd.buffers import RolloutAssembler


def make_step(eid, t, done=0.0, is_fir=None, obs_dim=4, n_act=2, hidden=8):
    return {
        "obs": np.full(obs_dim, float(t), dtype=np.float32),
        "act": np.array([t % n_act], dtype=np.float32),
        "rew": float(t),
        "logits": np.zeros(n_act, dtype=np.float32),
        "log_prob": np.array([-0.5], dtype=np.float32),
        "is_fir": is_fir if is_fir is not None else (1.0 if t == 0 else 0.0),
        "done": done,
        "hx": np.zeros(hidden, dtype=np.float32),
        "cx": np.zeros(hidden, dtype=np.float32),
        "id": eid,
    }


def run(coro):
    return asyncio.run(coro)


def test_full_trajectory_assembly_and_stacking():
    async def go():
        asm = RolloutAssembler(seq_len=5)
        for t in range(5):
            await asm.push(make_step("ep1", t))
        traj = await asm.pop()
        assert set(traj) >= {"obs", "act", "rew", "logits", "log_prob", "is_fir", "hx", "cx"}
        assert traj["obs"].shape == (5, 4)
        assert traj["rew"].shape == (5, 1)
        # steps in order
        torch.testing.assert_close(torch.as_tensor(traj["rew"]).squeeze(-1),
                                   torch.arange(5.0))
        assert asm.qsize() == 0

    run(go())


def test_interleaved_episodes_keyed_by_uuid():
    async def go():
        asm = RolloutAssembler(seq_len=3)
        # interleave two episodes' pushes
        for t in range(3):
            await asm.push(make_step("a", t))
            await asm.push(make_step("b", t + 10))
        ta = await asm.pop()
        tb = await asm.pop()
        got = {float(ta["rew"][0, 0]), float(tb["rew"][0, 0])}
        assert got == {0.0, 10.0}

    run(go())


def test_missing_field_asserts():
    async def go():
        asm = RolloutAssembler(seq_len=3)
        step = make_step("x", 0)
        del step["log_prob"]
        with pytest.raises(AssertionError):
            await asm.push(step)

    run(go())


def test_staleness_eviction():
    async def go():
        asm = RolloutAssembler(seq_len=5, stale_s=0.05)
        await asm.push(make_step("old", 0))
        await asyncio.sleep(0.1)
        # pushing a new episode evicts the stale partial
        for t in range(5):
            await asm.push(make_step("new", t + 100))
        traj = await asm.pop()
        assert float(traj["rew"][0, 0]) == 100.0
        assert "old" not in asm.active
        assert asm.qsize() == 0

    run(go())


def test_done_splice_sets_is_fir():
    async def go():
        asm = RolloutAssembler(seq_len=5)
        # 2-step episode that finishes early → parked
        await asm.push(make_step("short", 0))
        await asm.push(make_step("short", 1, done=1.0))
        assert len(asm.parked_done) == 1
        # new episode steps splice onto the parked trajectory
        for t in range(3):
            await asm.push(make_step("fresh", t, is_fir=1.0 if t == 0 else 0.0))
        traj = await asm.pop()
        assert traj["rew"].shape == (5, 1)
        # splice point (index 2) must have is_fir forced to 1.0
        assert float(traj["is_fir"][2, 0]) == 1.0
        assert float(traj["is_fir"][0, 0]) == 1.0
        assert float(traj["is_fir"][1, 0]) == 0.0

    run(go())


def test_splice_prefers_smallest_parked():
    async def go():
        asm = RolloutAssembler(seq_len=6)
        # interleave two episodes so both are active, then both finish short
        await asm.push(make_step("p3", 0))
        await asm.push(make_step("p2", 0))
        await asm.push(make_step("p2", 1, done=1.0))  # parked at len 2
        await asm.push(make_step("p3", 1))
        await asm.push(make_step("p3", 2, done=1.0))  # parked at len 3
        assert len(asm.parked_done) == 2
        # new episode should splice onto the SMALLEST parked trajectory (p2)
        await asm.push(make_step("fresh", 50))
        assert "p2" not in asm.parked_done  # the smaller one got used
        assert "p3" in asm.parked_done
        assert len(asm.active) == 1

    run(go())


def test_packed_chunk_splice_rewrites_is_fir_in_row():
    """Lean packed steps (wire.unpack_steps(lean=True)) through the splice:
    the parked-done trajectory absorbs the new episode's first step with
    is_fir forced to 1.0 INSIDE the stacked row (the fast stacking path
    reads rows, not the per-field dict)."""
    import numpy as np

    from pdrl_amd.buffers.wire import FIELD_ORDER, pack_steps, unpack_steps

    def mk(eid, t, done=0.0, is_fir=0.0):
        return {"obs": np.full(4, t, dtype=np.float32),
                "act": np.zeros(1, dtype=np.float32), "rew": float(t),
                "logits": np.zeros(2, dtype=np.float32),
                "log_prob": np.zeros(1, dtype=np.float32),
                "is_fir": is_fir, "done": done,
                "hx": np.zeros(64, dtype=np.float32),
                "cx": np.zeros(64, dtype=np.float32), "id": eid}

    async def go():
        asm = RolloutAssembler(seq_len=4)
        # episode A ends after 2 steps (done) → parked
        chunk_a = pack_steps([mk("A", 0, is_fir=1.0), mk("A", 1, done=1.0)])
        await asm.push_many(unpack_steps(chunk_a, lean=True))
        assert asm.qsize() == 0
        # episode B starts: first step splices onto parked A with is_fir=1
        chunk_b = pack_steps([mk("B", 10), mk("B", 11)])
        await asm.push_many(unpack_steps(chunk_b, lean=True))
        traj = await asm.pop()
        rew = np.asarray(traj["rew"]).reshape(-1)
        np.testing.assert_allclose(rew, [0.0, 1.0, 10.0, 11.0])
        fir = np.asarray(traj["is_fir"]).reshape(-1)
        np.testing.assert_allclose(fir, [1.0, 0.0, 1.0, 0.0])  # splice mark
        done = np.asarray(traj["done"]).reshape(-1)
        np.testing.assert_allclose(done, [0.0, 1.0, 0.0, 0.0])

    run(go())
