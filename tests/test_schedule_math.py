"""Property tests for the shared block/segment schedule math
(csrc/collectives/schedule.h): every (N, P, S) partitions [0, N)
exactly — the invariant both sides of every wire rely on."""
import pytest

import gloo_amd as ga


@pytest.mark.parametrize("n", [0, 1, 7, 100, 1001, 65536, 5_000_001])
@pytest.mark.parametrize("p", [1, 2, 3, 4, 7, 8])
def test_blocks_partition(n, p):
    cur = 0
    for b in range(p):
        off, ln = ga._C._block_of(n, p, b)
        assert off == cur
        cur += ln
    assert cur == n


@pytest.mark.parametrize("n", [0, 1, 100, 9999, 1 << 20])
@pytest.mark.parametrize("p", [2, 3, 8])
@pytest.mark.parametrize("s", [1, 2, 5])
def test_segments_partition_blocks(n, p, s):
    for b in range(p):
        boff, blen = ga._C._block_of(n, p, b)
        cur = boff
        for q in range(s):
            off, ln = ga._C._segment_of(n, p, b, q, s)
            assert off == cur
            cur += ln
        assert cur == boff + blen


@pytest.mark.parametrize("n", [0, 1, 17, 4096])
@pytest.mark.parametrize("base", [1, 2, 4])
def test_subspans_partition(n, base):
    cur = 0
    for j in range(base):
        off, ln = ga._C._subspan_of(0, n, j, base)
        assert off == cur
        cur += ln
    assert cur == n


def test_aligned_splits_partition_and_align():
    """Aligned variants: boundaries are multiples of A elements, pieces
    tile the range exactly, only the tail may be short."""
    import gloo_amd as ga

    for n in [1, 7, 1000, 1041667, 5_000_000, 100_000_000]:
        for P in [1, 2, 3, 7, 8]:
            for A in [1, 4, 8]:
                segs = [ga._C.block_of_a(n, P, b, A) for b in range(P)]
                total = 0
                for i, (off, ln) in enumerate(segs):
                    assert off % A == 0 or off == n
                    assert off == total
                    total += ln
                assert total == n
                # segments within block 0
                S = 5
                t2 = 0
                for s in range(S):
                    off, ln = ga._C.segment_of_a(n, P, 0, s, S, A)
                    assert off % A == 0 or off >= segs[0][1]
                    assert off == t2
                    t2 += ln
                assert t2 == segs[0][1]
