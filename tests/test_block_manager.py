"""Block-manager unit tests (the paged-KV accounting the router's KV% gauge
rides on — reference sim constants.py block math)."""

import pytest

from llm_instance_gateway_amd.engine.block_manager import BlockManager
from llm_instance_gateway_amd.engine.sequence import Sequence


def seq(n):
    return Sequence(list(range(1, n + 1)))


def test_grow_free_roundtrip_and_usage():
    bm = BlockManager(num_blocks=10, block_size=16)
    s = seq(5)
    bm.grow(s, 17)  # 17 tokens -> 2 blocks
    assert len(s.block_table) == 2
    assert bm.num_used_blocks == 2
    assert bm.usage == pytest.approx(0.2)
    bm.grow(s, 33)  # 3 blocks total -> +1
    assert len(s.block_table) == 3
    bm.free(s)
    assert bm.num_free_blocks == 10
    assert s.block_table == []
    assert bm.usage == 0.0


def test_grow_is_incremental_not_double_counted():
    bm = BlockManager(4, 16)
    s = seq(1)
    bm.grow(s, 16)
    bm.grow(s, 16)  # same target: no extra block
    assert len(s.block_table) == 1


def test_watermark_reserved_for_admission_only():
    bm = BlockManager(4, 16, watermark_blocks=2)
    s = seq(1)
    # admission path respects the watermark...
    assert bm.can_grow(s, 16 * 3, respect_watermark=True) is False
    assert bm.can_grow(s, 16 * 2, respect_watermark=True) is True
    # ...running decode growth may dip into it
    assert bm.can_grow(s, 16 * 4, respect_watermark=False) is True


def test_out_of_blocks_raises():
    bm = BlockManager(2, 16)
    s = seq(1)
    with pytest.raises(RuntimeError):
        bm.grow(s, 16 * 3)
    # failed grow must not have leaked partial allocations
    assert bm.num_free_blocks == 2
    assert s.block_table == []


def test_lifo_reuse_for_cache_locality():
    bm = BlockManager(8, 16)
    a, b = seq(1), seq(1)
    bm.grow(a, 16)
    first = a.block_table[0]
    bm.free(a)
    bm.grow(b, 16)
    assert b.block_table[0] == first  # most-recently-freed block reused
