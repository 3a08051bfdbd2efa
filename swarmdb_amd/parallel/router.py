"""Cross-GPU message routing — RCCL all-to-all over xGMI.

Agents are sharded across the node's GPUs (one process per GPU,
``torch.distributed`` with the nccl backend = RCCL on ROCm); each tick,
every rank partitions its outbound batch by destination rank and the
node exchanges (records, payloads) with a single
``all_to_all_single`` — direct point-to-point traffic over the
fully-connected xGMI mesh (7 links x ~153 GB/s per GPU), which is exactly
the shape all-to-all wants (no ring serialization). Broadcast messages
are replicated to every rank; each rank fans out to its locally resident
agents only.

This replaces the reference's single shared Kafka broker as the
cross-process transport (SURVEY.md §2.4 rows "partitioned topic" /
"cross-process shared state"; BASELINE config 4).

Works on CPU tensors with the gloo backend for CI without GPUs.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from ..runtime.engine import BROADCAST, REC_DTYPE

REC_BYTES = REC_DTYPE.itemsize  # 48


def shard_of(receiver_idx: np.ndarray, world_size: int) -> np.ndarray:
    """Owner rank of each (global) receiver index. Round-robin keeps
    every rank's agent set contiguous-in-modulus and load-balanced; the
    facade's string-keyed path uses the stable FNV hash instead
    (utils/hashing.shard_for)."""
    return receiver_idx % world_size


class GpuDirectRouter:
    """GPU-direct all-to-all routing: payloads never touch the host.

    Per tick: the local batch's payload is staged H2D once and a pack
    kernel scatters instances (broadcasts replicated per rank) into a
    torch-allocated device send buffer; RCCL `all_to_all_single` moves
    the sections point-to-point over xGMI; ingestion enqueues straight
    from the received device buffer (`enqueue_from_ptrs` — no D2H, no
    re-staging). The host only computes the instance permutation
    (vectorized numpy over 48-B records).
    """

    def __init__(self, engine, device: torch.device,
                 group: Optional[object] = None,
                 force_exchange: bool = False):
        self.engine = engine
        self.device = device
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        # force_exchange runs the full pack-kernel -> RCCL all_to_all ->
        # device-ingest pipeline even at world 1 (self-exchange) so a
        # 1-GPU box can execute and test the exact multi-GPU code path
        self.force_exchange = force_exchange
        # keep recv buffers alive while async ingest kernels may still
        # read them (2 ticks deep)
        self._hold: list = []

    def route_and_enqueue(self, recs: np.ndarray, payloads: bytes,
                          owner_of: Optional[np.ndarray] = None) -> int:
        W = self.world
        q = self.engine.q
        if W == 1 and not self.force_exchange:
            self.engine.enqueue_batch(recs, payloads)
            return len(recs)

        recv_field = recs["receiver"]
        bmask = recv_field == BROADCAST
        if owner_of is None:
            dest = np.where(
                bmask, 0, recv_field % np.uint32(W)
            ).astype(np.int64)
        else:
            dest = np.where(
                bmask, 0, owner_of[np.where(bmask, 0, recv_field)]
            ).astype(np.int64)

        # instance table: one per p2p message, W per broadcast
        p2p_idx = np.flatnonzero(~bmask)
        b_idx = np.flatnonzero(bmask)
        inst_msg = np.concatenate(
            [p2p_idx, np.repeat(b_idx, W)]
        ).astype(np.int64)
        inst_rank = np.concatenate(
            [dest[p2p_idx], np.tile(np.arange(W, dtype=np.int64), len(b_idx))]
        )
        order = np.argsort(inst_rank, kind="stable")
        inst_msg, inst_rank = inst_msg[order], inst_rank[order]
        n_inst = len(inst_msg)

        lens = recs["payload_len"][inst_msg].astype(np.uint32)
        lens16 = ((lens.astype(np.int64) + 15) // 16) * 16
        dst_off = np.zeros(n_inst, dtype=np.int64)
        np.cumsum(lens16[:-1], out=dst_off[1:])
        src_off = recs["payload_off"][inst_msg].astype(np.uint64)

        counts = np.bincount(inst_rank, minlength=W).astype(np.int64)
        bytes_per_rank = np.bincount(
            inst_rank, weights=lens16.astype(np.float64), minlength=W
        ).astype(np.int64)
        rank_base = np.zeros(W, dtype=np.int64)
        np.cumsum(bytes_per_rank[:-1], out=rank_base[1:])

        out_recs = recs[inst_msg].copy()
        out_recs["payload_off"] = (dst_off - rank_base[inst_rank]).astype(
            np.uint64
        )
        # the receive side ingests straight from device memory
        # (enqueue_from_ptrs bypasses the engine's staging transform), so
        # split bitmap handles -> (pool slot, epoch) here; the pool is
        # allocated in lockstep on every rank, so the mapping is global
        split = getattr(self.engine, "split_bitmap_handles", None)
        if split is not None:
            out_recs = split(out_recs)

        # exchange section sizes
        sz = torch.tensor(
            np.stack([counts, bytes_per_rank]).T.reshape(-1),
            dtype=torch.int64, device=self.device,
        )  # [c0,b0,c1,b1,...]
        in_sz = torch.empty_like(sz)
        dist.all_to_all_single(in_sz, sz, group=self.group)
        in_pairs = in_sz.cpu().numpy().reshape(W, 2)
        recv_counts, recv_bytes = in_pairs[:, 0], in_pairs[:, 1]

        # payload exchange: pack on-device, move over xGMI
        total_out = int(bytes_per_rank.sum())
        send_pay = torch.empty(total_out, dtype=torch.uint8,
                               device=self.device)
        if total_out:
            q.pack_exchange(
                np.frombuffer(payloads, dtype=np.uint8),
                src_off,
                dst_off.astype(np.uint64),
                lens,
                send_pay.data_ptr(),
            )
        recv_pay = torch.empty(int(recv_bytes.sum()), dtype=torch.uint8,
                               device=self.device)
        dist.all_to_all_single(
            recv_pay, send_pay,
            output_split_sizes=recv_bytes.tolist(),
            input_split_sizes=bytes_per_rank.tolist(),
            group=self.group,
        )

        # record exchange (48 B each)
        send_recs = torch.from_numpy(
            np.frombuffer(out_recs.tobytes(), dtype=np.uint8).copy()
        ).to(self.device)
        recv_recs = torch.empty(int(recv_counts.sum()) * REC_BYTES,
                                dtype=torch.uint8, device=self.device)
        dist.all_to_all_single(
            recv_recs, send_recs,
            output_split_sizes=(recv_counts * REC_BYTES).tolist(),
            input_split_sizes=(counts * REC_BYTES).tolist(),
            group=self.group,
        )
        torch.cuda.current_stream(self.device).synchronize()

        # ingest each source rank's section straight from device memory
        ingested = 0
        rec_off = 0
        pay_off = 0
        for r in range(W):
            n_r = int(recv_counts[r])
            if n_r:
                q.enqueue_from_ptrs(
                    recv_recs.data_ptr() + rec_off,
                    recv_pay.data_ptr() + pay_off,
                    n_r,
                )
                ingested += n_r
            rec_off += n_r * REC_BYTES
            pay_off += int(recv_bytes[r])
        self._hold.append((recv_recs, recv_pay))
        if len(self._hold) > 2:
            self._hold.pop(0)
        return ingested


class CrossGpuRouter:
    """One all-to-all exchange per tick.

    Wire format per destination rank: n records (48 B each, payload_off
    rebased to that destination's payload chunk) followed by the payload
    bytes. Counts go first in a small all-to-all so receive buffers can
    be sized exactly.
    """

    def __init__(self, device: torch.device, group: Optional[object] = None):
        self.device = device
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)

    def route(
        self, recs: np.ndarray, payloads: bytes,
        owner_of: Optional[np.ndarray] = None,
    ) -> Tuple[np.ndarray, bytes]:
        """Partition by destination, exchange, return this rank's inbound
        (records, payloads) — remote plus own, payload offsets rebased to
        the returned buffer. ``owner_of`` (agent idx -> rank, replicated)
        overrides the default modulo sharding — live re-sharding routes
        through it."""
        W = self.world
        recv = recs["receiver"]
        bmask = recv == BROADCAST
        if owner_of is None:
            dest = np.where(
                bmask, self.rank, recv % np.uint32(W)
            ).astype(np.int64)
        else:
            dest = np.where(
                bmask, self.rank, owner_of[np.where(bmask, 0, recv)]
            ).astype(np.int64)

        # build per-destination chunks (broadcasts replicated to all)
        src = np.frombuffer(payloads, dtype=np.uint8)
        all_lens = recs["payload_len"].astype(np.int64)
        all_offs = recs["payload_off"].astype(np.int64)
        # fast path: uniform 16-B-aligned stride (the batch hot path) —
        # payload gather becomes a vectorized 2-D fancy index
        stride = int(((all_lens.max() if len(recs) else 0) + 15) // 16 * 16)
        uniform = (
            len(recs) > 0
            and (all_lens == all_lens[0]).all()
            and (all_offs == np.arange(len(recs), dtype=np.int64) * stride).all()
            and len(src) >= len(recs) * stride
        )
        out_blobs = []
        for d in range(W):
            sel = (dest == d) | bmask
            sub = recs[sel].copy()
            if uniform:
                pay_bytes = (
                    src[: len(recs) * stride]
                    .reshape(len(recs), stride)[sel]
                    .tobytes()
                )
                sub["payload_off"] = (
                    np.arange(len(sub), dtype=np.uint64) * np.uint64(stride)
                )
            else:
                lens16 = ((sub["payload_len"].astype(np.int64) + 15) // 16) * 16
                offs = np.zeros(len(sub), dtype=np.int64)
                np.cumsum(lens16[:-1], out=offs[1:])
                dst = np.zeros(int(lens16.sum()), dtype=np.uint8)
                for i in range(len(sub)):
                    o, l, no = (
                        int(sub["payload_off"][i]),
                        int(sub["payload_len"][i]),
                        int(offs[i]),
                    )
                    dst[no : no + l] = src[o : o + l]
                sub["payload_off"] = offs.astype(np.uint64)
                pay_bytes = dst.tobytes()
            out_blobs.append(sub.tobytes() + pay_bytes)

        # one fused header exchange: (blob bytes, msg count) per rank
        hdr = torch.tensor(
            [
                [len(out_blobs[d]), int(((dest == d) | bmask).sum())]
                for d in range(W)
            ],
            dtype=torch.int64,
            device=self.device,
        ).reshape(-1)
        in_hdr = torch.empty_like(hdr)
        dist.all_to_all_single(in_hdr, hdr, group=self.group)
        in_pairs = in_hdr.reshape(W, 2).cpu()
        in_sizes, in_nmsgs = in_pairs[:, 0], in_pairs[:, 1]

        blob = b"".join(out_blobs)
        if blob:
            send_buf = torch.frombuffer(
                bytearray(blob), dtype=torch.uint8
            ).to(self.device, non_blocking=False)
        else:
            send_buf = torch.empty(0, dtype=torch.uint8, device=self.device)
        total_in = int(in_sizes.sum().item())
        recv_buf = torch.empty(total_in, dtype=torch.uint8, device=self.device)
        dist.all_to_all_single(
            recv_buf,
            send_buf,
            output_split_sizes=in_sizes.tolist(),
            input_split_sizes=[len(b) for b in out_blobs],
            group=self.group,
        )
        raw = recv_buf.cpu().numpy().tobytes()

        # unpack: concatenate records, rebase payload offsets
        rec_parts, pay_parts = [], []
        pay_base = 0
        off = 0
        for d in range(W):
            sz = int(in_sizes[d].item())
            nm = int(in_nmsgs[d].item())
            blob = raw[off : off + sz]
            off += sz
            rr = np.frombuffer(
                blob[: nm * REC_BYTES], dtype=REC_DTYPE, count=nm
            ).copy()
            pp = blob[nm * REC_BYTES :]
            rr["payload_off"] += np.uint64(pay_base)
            pay_base += len(pp)
            rec_parts.append(rr)
            pay_parts.append(pp)
        all_recs = (
            np.concatenate(rec_parts)
            if rec_parts
            else np.empty(0, dtype=REC_DTYPE)
        )
        return all_recs, b"".join(pay_parts)
