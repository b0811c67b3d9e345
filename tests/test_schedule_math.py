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
