"""Differential fuzz: the C++ block manager (csrc/block_manager.cpp) must
match the python PyBlockManager op-for-op — same tables, same slot lists,
same failures — across random add/append/pop/share/free traces."""
import random

import pytest
import torch  # noqa: F401 — loads libc10 for the extension

from agentcontrolplane_amd.engine.kv import OutOfBlocksError, PyBlockManager

try:
    from agentcontrolplane_amd import _C
    HAVE_NATIVE = hasattr(_C, "BlockManager")
except ImportError:
    HAVE_NATIVE = False

pytestmark = pytest.mark.skipif(not HAVE_NATIVE, reason="native extension not built")


def snapshot(bm, seqs):
    out = {"free": bm.free_blocks, "used": bm.used_blocks}
    for sid in seqs:
        if bm.has_seq(sid):
            out[sid] = (bm.seq_len(sid), list(bm.block_table(sid)))
    return out


@pytest.mark.parametrize("trial", range(6))
def test_differential_random_ops(trial):
    rng = random.Random(500 + trial)
    nb, bs = rng.choice([(24, 4), (64, 8), (40, 16)])
    py, cc = PyBlockManager(nb, bs), _C.BlockManager(nb, bs)
    live = set()
    next_id = 1
    for step in range(400):
        op = rng.random()
        if op < 0.2 or not live:
            sid = next_id
            next_id += 1
            py.add_seq(sid)
            cc.add_seq(sid)
            live.add(sid)
        elif op < 0.6:
            sid = rng.choice(sorted(live))
            n = rng.randrange(1, 2 * bs)
            pe = ce = ps = cs = None
            try:
                ps = py.append_tokens(sid, n)
            except OutOfBlocksError as e:
                pe = type(e).__name__
            try:
                cs = cc.append_tokens(sid, n)
            except Exception as e:
                ce = type(e).__name__
            assert (pe is None) == (ce is None), (step, pe, ce)
            if ps is not None:
                assert list(ps) == list(cs), step
        elif op < 0.7:
            sid = rng.choice(sorted(live))
            if py.seq_len(sid) > 0:
                pe = ce = None
                try:
                    py.pop_last_token(sid)
                except Exception as e:
                    pe = type(e).__name__
                try:
                    cc.pop_last_token(sid)
                except Exception as e:
                    ce = type(e).__name__
                assert pe == ce, (step, pe, ce)
        elif op < 0.8:
            # a fresh sequence adopts a donor's block-aligned prefix
            donor = rng.choice(sorted(live))
            blocks = min(len(py.block_table(donor)), rng.randrange(1, 4))
            if blocks > 0 and py.seq_len(donor) >= blocks * bs:
                sid = next_id
                next_id += 1
                # share_prefix registers the new sequence itself
                py.share_prefix(sid, donor, blocks, blocks * bs)
                cc.share_prefix(sid, donor, blocks, blocks * bs)
                live.add(sid)
        else:
            sid = rng.choice(sorted(live))
            py.free_seq(sid)
            cc.free_seq(sid)
            live.discard(sid)
        assert snapshot(py, sorted(live)) == snapshot(cc, sorted(live)), step
