"""Sequence-parallel re-sharding (Ulysses-style) composed from AlltoAll.

The reference's model-group AlltoAll (its case-4/5 "switch which dimension
is sharded" transition) is structurally the modern
sequence<->head re-shard; this module packages it:

    seq->head : [B, S/P, H]  ->  [B, S, H/P]
    head->seq : [B, S, H/P]  ->  [B, S/P, H]

Works on numpy arrays or torch tensors (contiguous, fp32); P = the model
group size of `dist`.
"""
import numpy as np

import mlsl_amd as mx


def _xp(t):
    return np if isinstance(t, np.ndarray) else __import__("torch")


def seq_to_head(dist, x, group="model"):
    """x: [B, S_local, H] with S_local = S/P -> [B, S, H/P]."""
    P = dist.process_count(group)
    if P == 1:
        return x
    xp = _xp(x)
    B, S_local, H = x.shape
    assert H % P == 0, "H must divide by the group size"
    Hl = H // P
    # send block p = x[:, :, p*Hl:(p+1)*Hl] flattened
    if xp is np:
        send = np.ascontiguousarray(
            np.stack([x[:, :, p * Hl:(p + 1) * Hl] for p in range(P)]))
        recv = np.empty_like(send)
    else:
        send = xp.stack([x[:, :, p * Hl:(p + 1) * Hl] for p in range(P)]).contiguous()
        recv = xp.empty_like(send)
    per = B * S_local * Hl
    mx.wait(dist.all_to_all(send, per, recv, dtype="f32", group=group))
    # recv[p] = peer p's sequence slice of my head slice
    if xp is np:
        return np.concatenate([recv[p] for p in range(P)], axis=1)
    return xp.cat([recv[p] for p in range(P)], dim=1)


def head_to_seq(dist, x, group="model"):
    """x: [B, S, H_local] with H_local = H/P -> [B, S/P, H]."""
    P = dist.process_count(group)
    if P == 1:
        return x
    xp = _xp(x)
    B, S, Hl = x.shape
    assert S % P == 0, "S must divide by the group size"
    Sl = S // P
    if xp is np:
        send = np.ascontiguousarray(
            np.stack([x[:, p * Sl:(p + 1) * Sl, :] for p in range(P)]))
        recv = np.empty_like(send)
    else:
        send = xp.stack([x[:, p * Sl:(p + 1) * Sl, :] for p in range(P)]).contiguous()
        recv = xp.empty_like(send)
    per = B * Sl * Hl
    mx.wait(dist.all_to_all(send, per, recv, dtype="f32", group=group))
    if xp is np:
        return np.concatenate([recv[p] for p in range(P)], axis=2)
    return xp.cat([recv[p] for p in range(P)], dim=2)


def ring_exchange(dist, block, group="model"):
    """One ring-attention-style neighbor step: send my KV block to the next
    rank, receive the previous rank's (SendRecvList is exposed C++-side;
    composed here from the same pairwise machinery)."""
    P = dist.process_count(group)
    if P == 1:
        return block
    xp = _xp(block)
    r = dist.process_idx(group)
    flat = block.reshape(-1)
    n = flat.shape[0]
    # alltoallv with only two non-empty lanes = the neighbor exchange
    scnt = [0] * P
    rcnt = [0] * P
    soff = [0] * P
    roff = [0] * P
    scnt[(r + 1) % P] = n
    rcnt[(r - 1 + P) % P] = n
    out = xp.empty_like(flat)
    mx.wait(dist.all_to_allv(flat, scnt, soff, out, rcnt, roff,
                             dtype="f32", group=group))
    return out.reshape(block.shape)
