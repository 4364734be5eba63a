"""Shared distributed panel-broadcast helpers for the BLAS-3 algorithms.

These implement the reference's panel broadcast building blocks
(``communication/broadcast_panel.h:45-63`` plain and ``:78-189`` transposed):

* a COLUMN panel of matrix column ``k`` (tiles A[i,k] keyed by local tile-row)
  broadcast along the ROW direction so every rank column holds the tiles for
  its local rows;
* a ROW panel of matrix row ``k`` (tiles A[k,j] keyed by local tile-col)
  broadcast along the COLUMN direction;
* the two TRANSPOSED conversions (col panel -> row panel and back): tile
  A[j,k] is re-keyed from "local row of j" to "local col of j" (or vice
  versa) by a local copy on the owning rank row/col followed by a broadcast
  along the orthogonal direction — the two-hop scheme of the reference.

Ordering: every rank of a process group issues the group's collectives in the
same deterministic loop order (ranks sharing a row group have identical local
column counts and vice versa), which is the RCCL equivalent of the reference's
``CommunicatorPipeline::exclusive()`` ordering guarantee.
"""

from __future__ import annotations

from ..comm import collectives as coll


def bcast_col_panel(mat, grid, k: int, li0: int, li1: int, panel) -> None:
    """panel.slot(li) <- A[gi(li), k] for li in [li0, li1), on every rank col.

    The owner column of matrix column ``k`` fills the slots from its local
    tiles; one contiguous broadcast along the row direction replicates them.
    """
    d = mat.dist
    if li1 <= li0:
        return
    kc = d.rank_of_tile_col(k)
    if d.rank_col == kc:
        for li in range(li0, li1):
            i = d.global_tile_of_local((li, 0))[0]
            panel.slot(li).copy_(mat.tile((i, k)))
    if grid.row_group is not None:
        coll.broadcast(panel.range_view(li0, li1), grid.global_rank_of_row_member(kc), grid.row_group)


def bcast_row_panel(mat, grid, k: int, lj0: int, lj1: int, panel) -> None:
    """panel.slot(lj) <- A[k, gj(lj)] for lj in [lj0, lj1), on every rank row."""
    d = mat.dist
    if lj1 <= lj0:
        return
    kr = d.rank_of_tile_row(k)
    if d.rank_row == kr:
        for lj in range(lj0, lj1):
            j = d.global_tile_of_local((0, lj))[1]
            panel.slot(lj).copy_(mat.tile((k, j)))
    if grid.col_group is not None:
        coll.broadcast(panel.range_view(lj0, lj1), grid.global_rank_of_col_member(kr), grid.col_group)


def transpose_col_to_row(dist, grid, col_panel, row_panel, lj0: int, lj1: int) -> None:
    """row_panel.slot(lj) <- col_panel content for global index gj(lj).

    Precondition: ``col_panel`` already holds tile ``j`` at the local-row slot
    of ``j`` on every rank of the grid row owning tile-row ``j`` (e.g. after
    :func:`bcast_col_panel`). Per slot: local copy on the owning rank row,
    then a broadcast along the column direction.
    """
    d = dist
    for lj in range(lj0, lj1):
        j = d.global_tile_of_local((0, lj))[1]
        jr = d.rank_of_tile_row(j)
        if d.rank_row == jr:
            lji = d.next_local_tile_row(j)
            row_panel.slot(lj).copy_(col_panel.slot(lji))
        if grid.col_group is not None:
            coll.broadcast(row_panel.slot(lj), grid.global_rank_of_col_member(jr), grid.col_group)


def transpose_row_to_col(dist, grid, row_panel, col_panel, li0: int, li1: int) -> None:
    """col_panel.slot(li) <- row_panel content for global index gi(li).

    Precondition: ``row_panel`` holds tile ``i`` at the local-col slot of ``i``
    on every rank of the grid column owning tile-col ``i`` (e.g. after
    :func:`bcast_row_panel`).
    """
    d = dist
    for li in range(li0, li1):
        i = d.global_tile_of_local((li, 0))[0]
        ic = d.rank_of_tile_col(i)
        if d.rank_col == ic:
            lij = d.next_local_tile_col(i)
            col_panel.slot(li).copy_(row_panel.slot(lij))
        if grid.row_group is not None:
            coll.broadcast(col_panel.slot(li), grid.global_rank_of_row_member(ic), grid.row_group)


def bcast_diag_to_row(mat, grid, k: int, ws):
    """Return A[k,k] on every rank of the grid row owning tile-row k (others: None).

    ``ws`` is a tile-sized workspace used on non-owner columns.
    """
    d = mat.dist
    kr, kc = d.rank_of_tile(k if isinstance(k, tuple) else (k, k))
    if d.rank_row != kr:
        return None
    diag = mat.tile((k, k)) if d.rank_col == kc else ws
    if grid.row_group is not None:
        coll.broadcast(diag, grid.global_rank_of_row_member(kc), grid.row_group)
    return diag


def bcast_diag_to_col(mat, grid, k: int, ws):
    """Return A[k,k] on every rank of the grid column owning tile-col k."""
    d = mat.dist
    kr, kc = d.rank_of_tile((k, k))
    if d.rank_col != kc:
        return None
    diag = mat.tile((k, k)) if d.rank_row == kr else ws
    if grid.col_group is not None:
        coll.broadcast(diag, grid.global_rank_of_col_member(kr), grid.col_group)
    return diag


def bcast_diag_to_all(mat, grid, k: int, ws):
    """Return A[k,k] replicated on every rank (two-hop: row then col)."""
    d = mat.dist
    kr, kc = d.rank_of_tile((k, k))
    diag = mat.tile((k, k)) if (d.rank_row, d.rank_col) == (kr, kc) else ws
    if d.rank_row == kr and grid.row_group is not None:
        coll.broadcast(diag, grid.global_rank_of_row_member(kc), grid.row_group)
    if grid.col_group is not None:
        coll.broadcast(diag, grid.global_rank_of_col_member(kr), grid.col_group)
    return diag
