"""CPU unit tests for the min-skyline memory packer (schedule/)."""
import random

from easydist_amd.schedule.efficient_memory_scheduler import (ALIGN,
                                                              pack_buffers)
from easydist_amd.schedule.lifetime import Buffer


def _overlap_time(a: Buffer, b: Buffer) -> bool:
    return not (a.end < b.start or b.end < a.start)


def _overlap_addr(oa, a: Buffer, ob, b: Buffer) -> bool:
    return not (oa + a.size <= ob or ob + b.size <= oa)


def test_pack_disjoint_lifetimes_share_memory():
    bufs = [Buffer("a", 0, 1024, 0, 1, False),
            Buffer("b", 0, 1024, 2, 3, False)]
    addr, peak = pack_buffers(bufs)
    assert peak == 1024           # reuse: both at offset 0
    assert addr[("a", 0)] == addr[("b", 0)] == 0


def test_pack_overlapping_lifetimes_disjoint_addresses():
    bufs = [Buffer("a", 0, 1000, 0, 5, False),
            Buffer("b", 0, 2000, 3, 8, False),
            Buffer("c", 0, 500, 4, 4, True)]
    addr, peak = pack_buffers(bufs)
    for i, x in enumerate(bufs):
        for y in bufs[i + 1:]:
            if _overlap_time(x, y):
                assert not _overlap_addr(addr[(x.node_name, 0)], x,
                                         addr[(y.node_name, 0)], y), (x, y)
    assert peak <= sum(b.size for b in bufs) + 2 * ALIGN


def test_pack_random_no_overlap_and_saves():
    rng = random.Random(0)
    bufs = []
    for i in range(200):
        start = rng.randrange(0, 100)
        end = start + rng.randrange(0, 10)
        bufs.append(Buffer(f"n{i}", 0, rng.choice([256, 1024, 4096, 100000]),
                           start, end, False))
    addr, peak = pack_buffers(bufs)
    for i, x in enumerate(bufs):
        for y in bufs[i + 1:]:
            if _overlap_time(x, y):
                assert not _overlap_addr(addr[(x.node_name, 0)], x,
                                         addr[(y.node_name, 0)], y)
    naive = sum(b.size for b in bufs)
    assert peak < naive * 0.6     # interleaved lifetimes must reuse memory


def test_ilp_pack_at_least_as_good_as_skyline():
    from easydist_amd.schedule.ilp_memory_scheduler import ilp_pack_buffers
    bufs = [Buffer("a", 0, 1024, 0, 3, False),
            Buffer("b", 0, 1024, 1, 2, False),
            Buffer("c", 0, 2048, 2, 5, False),
            Buffer("d", 0, 512, 4, 6, False),
            Buffer("e", 0, 1024, 6, 8, False)]
    addr_h, peak_h = pack_buffers(bufs)
    addr_i, peak_i = ilp_pack_buffers(bufs)
    assert peak_i <= peak_h
    for i, x in enumerate(bufs):
        for y in bufs[i + 1:]:
            if _overlap_time(x, y):
                assert not _overlap_addr(addr_i[(x.node_name, 0)], x,
                                         addr_i[(y.node_name, 0)], y)


def test_plot_plan(tmp_path):
    from easydist_amd.schedule.efficient_memory_scheduler import (
        pack_buffers, plot_plan)
    from easydist_amd.schedule.lifetime import Buffer
    bufs = [Buffer(node_name=f"b{i}", alloc_idx=0,
                   size=(i % 5 + 1) * 1024, start=i, end=i + 4,
                   is_temp=False) for i in range(20)]
    addr, peak = pack_buffers(bufs)
    out = tmp_path / "plan.png"
    plot_plan(bufs, addr, peak, str(out))
    assert out.stat().st_size > 1000
