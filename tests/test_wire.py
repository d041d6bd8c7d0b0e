"""Packed rollout wire format (pdrl_amd/buffers/wire.py)."""
import numpy as np
import pytest

from pdrl_amd.buffers.wire import FIELD_ORDER, is_packed, pack_steps, unpack_steps


def _mk_step(eid, f=4, a=1, l=2, h=8):
    rng = np.random.default_rng(hash(eid) % 2**32)
    return {
        "obs": rng.standard_normal(f).astype(np.float32),
        "act": rng.standard_normal(a).astype(np.float32),
        "rew": float(rng.standard_normal()),
        "logits": rng.standard_normal(l).astype(np.float32),
        "log_prob": rng.standard_normal(1).astype(np.float32),
        "is_fir": 1.0,
        "done": 0.0,
        "hx": rng.standard_normal(h).astype(np.float32),
        "cx": rng.standard_normal(h).astype(np.float32),
        "id": eid,
    }


def test_pack_unpack_roundtrip():
    steps = [_mk_step(f"e{i % 3}") for i in range(7)]
    packed = pack_steps(steps)
    assert is_packed(packed) and not is_packed(steps) and not is_packed(steps[0])
    assert packed["pk"].dtype == np.float32 and packed["pk"].shape[0] == 7
    out = unpack_steps(packed)
    assert len(out) == 7
    for s0, s1 in zip(steps, out):
        assert s1["id"] == s0["id"]
        for k in FIELD_ORDER:
            np.testing.assert_allclose(
                np.asarray(s1[k], dtype=np.float32).reshape(-1),
                np.asarray(s0[k], dtype=np.float32).reshape(-1),
                rtol=0, atol=0)


def test_unpacked_views_share_chunk_memory():
    steps = [_mk_step("x") for _ in range(3)]
    packed = pack_steps(steps)
    out = unpack_steps(packed)
    assert out[0]["obs"].base is packed["pk"]  # zero-copy views


def test_mixed_uuid_chunk_assembles(params):
    """A packed chunk with interleaved uuids routes per step by id."""
    import asyncio

    from pdrl_amd.buffers import RolloutAssembler

    S = 5
    steps = []
    for t in range(S):
        for eid in ("aaa", "bbb"):
            s = _mk_step(eid)
            s["is_fir"] = 1.0 if t == 0 else 0.0
            steps.append(s)
    out = unpack_steps(pack_steps(steps))

    async def run():
        asm = RolloutAssembler(S, stale_s=1e9)
        for s in out:
            await asm.push(s)
        got = []
        while not asm.out_queue.empty():
            got.append(asm.out_queue.get_nowait())
        return got

    trajs = asyncio.run(run())
    assert len(trajs) == 2
    for tr in trajs:
        assert tr["obs"].shape == (S, 4)


def test_packed_weights_roundtrip():
    """WeightPublisher payload → encode/decode → unpack → load_state_dict
    reproduces the exact actor weights (the learner's fast broadcast path)."""
    import torch

    from pdrl_amd.agents.learner import WeightPublisher
    from pdrl_amd.buffers.wire import is_packed_weights, unpack_weights
    from pdrl_amd.networks import MlpLSTMSingle
    from pdrl_amd.utils import Protocol, decode, encode

    torch.manual_seed(4)
    src = MlpLSTMSingle(4, 2, 5, 64)
    dst = MlpLSTMSingle(4, 2, 5, 64)
    pub = WeightPublisher(src.actor, "cpu")
    header, payload = encode(Protocol.Model, pub.payload(), compress=False)
    proto, obj = decode(header, payload)
    assert proto is Protocol.Model and is_packed_weights(obj)
    dst.actor.load_state_dict(unpack_weights(obj))
    for a, b in zip(src.actor.parameters(), dst.actor.parameters()):
        torch.testing.assert_close(a, b)


def test_weight_publisher_begin_finish_cpu_roundtrip():
    """WeightPublisher.begin()/finish() (the staged-mode pipelined split)
    on CPU: payload unpacks back into the exact state_dict."""
    import torch

    from pdrl_amd.agents.learner import WeightPublisher
    from pdrl_amd.buffers.wire import is_packed_weights, unpack_weights
    from pdrl_amd.networks import MlpLSTMSingle

    torch.manual_seed(0)
    m = MlpLSTMSingle(4, 2, 5, 64)
    pub = WeightPublisher(m.actor, "cpu")
    pub.begin()  # no-op on CPU, must not raise
    payload = pub.finish()
    assert is_packed_weights(payload)
    sd = unpack_weights(payload)
    ref = m.actor.state_dict()
    assert set(sd) == set(ref)
    for k in ref:
        torch.testing.assert_close(sd[k], ref[k])
