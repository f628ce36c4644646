"""Pin the fitted routing heuristics so shape-dependent dispatch cannot
silently regress.

Three routing decisions are pinned across a (B, H, Tq) grid:
  * ops.flash.decode_loop_chunks  — spec-decode loop vs single dispatch
    (crossover measured at Tq~256 for H=32, profiles/decode_matrix_1gpu.jsonl);
  * parallel.tree.default_q_chunk — prefill chunk size (grid-fill rule);
  * parallel.combine.auto_strategy — allgather vs allreduce payload rule.
"""

import pytest

from tree_attention_torch_amd.ops.flash import decode_loop_chunks
from tree_attention_torch_amd.parallel.combine import auto_strategy
from tree_attention_torch_amd.parallel.tree import default_q_chunk


class TestDecodeLoopRouting:
    def test_plain_decode_single_dispatch(self):
        # Tq*G <= 16 fits one decode tile: extension dispatch, no loop.
        assert decode_loop_chunks(tq=1, g=1, b=1, hq=32) == 0
        assert decode_loop_chunks(tq=16, g=1, b=1, hq=32) == 0
        assert decode_loop_chunks(tq=2, g=8, b=1, hq=32) == 0
        assert decode_loop_chunks(tq=1, g=16, b=1, hq=16) == 0

    def test_spec_decode_loops(self):
        # Tq just past one tile at H=32: loop the decode kernel.
        assert decode_loop_chunks(tq=17, g=1, b=1, hq=32) == 2
        assert decode_loop_chunks(tq=32, g=1, b=1, hq=32) == 2
        assert decode_loop_chunks(tq=64, g=1, b=1, hq=32) == 4
        assert decode_loop_chunks(tq=128, g=1, b=1, hq=32) == 8

    def test_crossover_at_tq256_h32(self):
        # measured crossover: n_chunks == 512/blocks is the tie, strict <
        # keeps the loop only below it (prefill 1.47 vs loop 1.61 ms at 32K).
        assert decode_loop_chunks(tq=240, g=1, b=1, hq=32) == 15
        assert decode_loop_chunks(tq=256, g=1, b=1, hq=32) == 0  # prefill

    def test_prefill_sized_batches_never_loop(self):
        for tq in (512, 4096, 32768):
            assert decode_loop_chunks(tq=tq, g=1, b=1, hq=32) == 0

    def test_gqa_spec_decode(self):
        # G=8: two query rows per tile; Tq=4 -> 2 chunks.
        assert decode_loop_chunks(tq=4, g=8, b=1, hq=32) == 2
        # G=16: one row per tile.
        assert decode_loop_chunks(tq=2, g=16, b=1, hq=16) == 2

    def test_small_head_count_loops_longer(self):
        # H=8: prefill grid is 8 blocks/256 rows -> max_chunks 64; the loop
        # should still be taken at Tq=128 (16 chunks).
        assert decode_loop_chunks(tq=128, g=1, b=1, hq=8) == 8
        assert decode_loop_chunks(tq=256, g=1, b=1, hq=8) == 16
        # Tq=512 doubles the prefill q-blocks (ceil(512/256) = 2), landing
        # exactly on the n_chunks == 512/blocks tie -> prefill.
        assert decode_loop_chunks(tq=512, g=1, b=1, hq=8) == 0

    def test_full_grid_never_loops(self):
        # B*Hq >= 512: prefill fills the chip at any Tq > 16.
        assert decode_loop_chunks(tq=32, g=1, b=16, hq=32) == 0

    def test_mqa_group_over_16_raises(self):
        # used to ZeroDivisionError before the TORCH_CHECK.
        with pytest.raises(ValueError, match="group size"):
            decode_loop_chunks(tq=1, g=32, b=1, hq=32)


class TestPrefillChunking:
    def test_small_tq_unchunked(self):
        assert default_q_chunk(1024, 1, 32) == 1024
        assert default_q_chunk(4096, 1, 32) == 4096

    def test_h32_chunk_is_4096(self):
        # 512 blocks * 256 rows / 32 heads = 4096 (measured optimal at 32K
        # AND 64K rows, H=32 — docs/ROUND1_NOTES.md).
        assert default_q_chunk(32768, 1, 32) == 4096

    def test_h8_chunk_is_16384(self):
        assert default_q_chunk(65536, 1, 8) == 16384

    def test_chunk_never_underfills_grid(self):
        for b, hq in [(1, 8), (1, 32), (2, 16), (4, 8)]:
            chunk = default_q_chunk(1 << 20, b, hq)
            blocks = b * hq * (chunk // 256)
            assert blocks >= 512, (b, hq, chunk, blocks)

    def test_floor_at_4096(self):
        # many heads: grid fills at small chunks, but chunks below 4096 rows
        # add launch overhead for nothing.
        assert default_q_chunk(1 << 20, 8, 64) == 4096


class TestCombineStrategy:
    def test_decode_payload_allgather(self):
        # B=1 H=32 Tq=1 D=128 fp32 = 16 KiB -> latency-bound.
        assert auto_strategy(1 * 32 * 1 * 128) == "allgather"

    def test_prefill_chunk_allreduce(self):
        # 4096-row H=32 chunk = 64 MiB fp32 -> bandwidth-bound.
        assert auto_strategy(1 * 32 * 4096 * 128) == "allreduce"

    def test_boundary(self):
        assert auto_strategy((1 << 18) - 1) == "allgather"
        assert auto_strategy(1 << 18) == "allreduce"


def test_mx_decode_shaped_pin():
    """MX route decision: G*Tq <= 16 -> hardware-scale decode kernel."""
    from tree_attention_torch_amd.ops.flash import mx_decode_shaped

    assert mx_decode_shaped(32, 32, 1)        # MHA decode
    assert mx_decode_shaped(8, 1, 2)          # GQA8, tq=2 (exactly 16)
    assert not mx_decode_shaped(8, 1, 3)      # 24 rows -> prefill
    assert not mx_decode_shaped(32, 32, 17)   # spec batch -> prefill
    assert mx_decode_shaped(16, 16, 16)       # 16 rows boundary


def test_mx_rows_contig_pin():
    """Zero-copy eligibility: session-cache views qualify, transposes and
    gathered tensors do not."""
    import torch

    from tree_attention_torch_amd.ops.flash import _rows_contig

    cache = torch.zeros(2, 4, 256, 128, dtype=torch.uint8)
    assert _rows_contig(cache)
    assert _rows_contig(cache[:, :, :192])          # sliced view (live len)
    assert not _rows_contig(cache.transpose(2, 3))  # rows not packed
    assert not _rows_contig(cache[:, :, :, :64])   # row prefix: holes
    scales = torch.zeros(2, 4, 256, 4, dtype=torch.uint8)
    assert _rows_contig(scales[:, :, :64])
