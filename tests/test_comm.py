"""Communication layer unit tests (reference test/unit/communication)."""
import torch
import torch.distributed as dist

from dlaf_amd import CommGrid, Matrix
from dlaf_amd.comm import collectives as coll
from dlaf_amd.matrix.panel import Panel
from dlaf_amd.core.distribution import Distribution
from dist_utils import run_distributed


def _grid_worker(rank, ws, gr, gc):
    g = CommGrid(gr, gc)
    # rank math (row-major order)
    assert g.rank_full(g.rank_row, g.rank_col) == rank
    assert g.rank_row == rank // gc and g.rank_col == rank % gc
    # row broadcast: row leader sends its rank; all in the row must agree
    t = torch.tensor([float(rank)])
    if g.row_group is not None:
        coll.broadcast(t, g.global_rank_of_row_member(0), g.row_group)
        assert int(t.item()) == g.rank_row * gc
    # col allreduce
    t = torch.tensor([1.0])
    if g.col_group is not None:
        coll.all_reduce_sum(t, g.col_group)
        assert int(t.item()) == gr
    # complex payloads through the real view
    c = torch.tensor([complex(rank, -rank)])
    coll.all_reduce_sum(c, g.full_group)
    s = sum(range(ws))
    assert c.item() == complex(s, -s)
    return True


def test_comm_grid_6ranks():
    assert all(run_distributed(_grid_worker, 6, args=(2, 3)))


def test_comm_grid_4ranks():
    assert all(run_distributed(_grid_worker, 4, args=(2, 2)))


def _p2p_worker(rank, ws):
    t = torch.zeros(4)
    if rank == 0:
        coll.send(torch.arange(4.0), 1, tag=7)
    elif rank == 1:
        coll.recv(t, 0, tag=7)
        assert torch.equal(t, torch.arange(4.0))
    return True


def test_p2p():
    assert all(run_distributed(_p2p_worker, 2))


def _panel_bcast_worker(rank, ws, gr, gc):
    from dlaf_amd.algs import _panels as pan
    from dlaf_amd.matrix import util as mutil
    g = CommGrid(gr, gc)
    mat = Matrix.create(16, 16, 4, 4, grid=g)
    mutil.set_random(mat, seed=3)
    a = mat.to_global()
    colp = Panel(Panel.COL, mat.dist, mat.dtype, mat.device)
    rowp = Panel(Panel.ROW, mat.dist, mat.dtype, mat.device)
    k = 1
    d = mat.dist
    lr = d.local_nr_tiles[0]
    li0 = d.next_local_tile_row(k + 1)
    pan.bcast_col_panel(mat, g, k, li0, lr, colp)
    # every rank must now hold A[i, k] for its local rows i > k
    for li in range(li0, lr):
        i = d.global_tile_of_local((li, 0))[0]
        want = a[i * 4:(i + 1) * 4, k * 4:(k + 1) * 4]
        assert torch.equal(colp.slot(li), want)
    lj0 = d.next_local_tile_col(k + 1)
    lc = d.local_nr_tiles[1]
    pan.transpose_col_to_row(d, g, colp, rowp, lj0, lc)
    for lj in range(lj0, lc):
        j = d.global_tile_of_local((0, lj))[1]
        want = a[j * 4:(j + 1) * 4, k * 4:(k + 1) * 4]
        assert torch.equal(rowp.slot(lj), want)
    return True


def test_panel_broadcasts():
    assert all(run_distributed(_panel_bcast_worker, 4, args=(2, 2)))
