"""Distributed QPager — one state-vector page per torch.distributed rank.

MI355X-native replacement for the reference's QPager multi-device layer
(/root/reference/src/qpager.cpp): one process per GPU,
RCCL over xGMI via torch.distributed for the cross-page half-exchanges
(`ShuffleBuffers`, opencl.cpp:254-264 in the reference staged through HOST
memory — here it is a direct GPU-to-GPU sendrecv on dlpack views of HBM).

Page-index tricks are preserved: X/Y/phase gates on meta qubits move no
data (page relabeling + per-page scalars, qpager.cpp:509-525); controlled
gates split into intra / semi-meta / meta classes (qpager.cpp:1011-1058).

All ranks hold the same `page_of_rank` table and the same decision RNG, so
control flow is replicated deterministically; only amplitude data moves.
"""

import numpy as np
import torch
import torch.distributed as dist

import qrack_amd as qa


class DistQPager:
    def __init__(self, qubits, precision="fp32", engine="hip", seed=1234, device_id=0):
        assert dist.is_initialized(), "torch.distributed must be initialized"
        self.world = dist.get_world_size()
        self.rank = dist.get_rank()
        assert self.world & (self.world - 1) == 0, "world size must be a power of 2"
        self.meta_bits = (self.world - 1).bit_length()
        self.num_qubits = qubits
        self.qpp = qubits - self.meta_bits  # qubits per page
        assert self.qpp >= 1
        self.precision = precision
        self.engine_kind = engine
        self.device_id = device_id
        seed = 1234 if seed is None or seed < 0 else seed
        self.q = qa.create_simulator(
            self.qpp, precision=precision, engine=engine, seed=seed, device_id=device_id
        )
        # page_of_rank[r] = logical page index rank r holds (identity at reset)
        self.page_of_rank = list(range(self.world))
        self.rng = np.random.default_rng(seed)  # replicated decision RNG
        self.torch_dtype = torch.complex64 if precision == "fp32" else torch.complex128
        self.page_len = 1 << self.qpp

    # ---- helpers ------------------------------------------------------------

    @property
    def my_page(self):
        return self.page_of_rank[self.rank]

    def _rank_of_page(self, page):
        return self.page_of_rank.index(page)

    def _is_hip(self):
        return self.engine_kind == "hip"

    def _sync_engine(self):
        self.q.finish()

    def _sync_torch(self):
        if self._is_hip():
            torch.cuda.synchronize(self.device_id)

    def _half_view(self, low_half):
        """torch view (zero-copy on HIP) of one half of the local page."""
        off = 0 if low_half else self.page_len // 2
        cap = self.q.dlpack_view(off, self.page_len // 2)
        return torch.from_dlpack(cap)

    def _shuffle(self, partner_rank, i_am_low):
        """Swap my (upper if low page else lower) half with the partner's
        opposite half — the reference's cross-device ShuffleBuffers
        (opencl.cpp:254-264, staged through HOST there), as an RCCL sendrecv
        pair over xGMI on zero-copy HBM views here. Non-NCCL backends (gloo
        CI without GPUs, or single-GPU validation) stage through host."""
        self._sync_engine()
        nccl = dist.get_backend() == "nccl"
        if nccl or not self._is_hip():
            view = self._half_view(low_half=not i_am_low)
            tmp = torch.empty_like(view)
            ops = [
                dist.P2POp(dist.isend, view, partner_rank),
                dist.P2POp(dist.irecv, tmp, partner_rank),
            ]
            reqs = dist.batch_isend_irecv(ops)
            for r in reqs:
                r.wait()
            view.copy_(tmp)
            self._sync_torch()
        else:
            off = self.page_len // 2 if i_am_low else 0
            buf = self.q.get_amplitude_page(off, self.page_len // 2)
            send = torch.from_numpy(buf)
            tmp = torch.empty_like(send)
            reqs = dist.batch_isend_irecv(
                [
                    dist.P2POp(dist.isend, send, partner_rank),
                    dist.P2POp(dist.irecv, tmp, partner_rank),
                ]
            )
            for r in reqs:
                r.wait()
            self.q.set_amplitude_page(tmp.numpy(), off)

    def _split_controls(self, controls):
        local = [c for c in controls if c < self.qpp]
        meta = [c - self.qpp for c in controls if c >= self.qpp]
        return local, meta

    def _meta_controls_satisfied(self, meta, anti=False):
        page = self.my_page
        for b in meta:
            bit = (page >> b) & 1
            if anti and bit:
                return False
            if not anti and not bit:
                return False
        return True

    def _pair_meta_controls_satisfied(self, meta, target_bit):
        # controls other than the target bit are shared by both pages of a pair
        page = self.my_page
        for b in meta:
            if b == target_bit:
                continue
            if not ((page >> b) & 1):
                return False
        return True

    # ---- state management ----------------------------------------------------

    def set_permutation(self, perm):
        self.page_of_rank = list(range(self.world))
        page = perm >> self.qpp
        if self.my_page == page:
            self.q.set_permutation(perm & (self.page_len - 1))
        else:
            self.q.zero_amplitudes()

    def finish(self):
        self._sync_engine()

    def get_state_vector(self):
        """Gather the full state on every rank (test helper; small widths)."""
        local = np.asarray(self.q.get_state_vector())
        out = [None] * self.world
        dist.all_gather_object(out, (self.my_page, local))
        full = np.zeros(1 << self.num_qubits, dtype=local.dtype)
        for page, arr in out:
            full[page * self.page_len : (page + 1) * self.page_len] = arr
        return full

    # ---- gates ---------------------------------------------------------------

    def mtrx(self, m, target):
        self.mcmtrx([], m, target)

    def phase(self, tl, br, target):
        self.mcphase([], tl, br, target)

    def invert(self, tr, bl, target):
        self.mcmtrx([], [0, tr, bl, 0], target)

    def mcmtrx(self, controls, m, target):
        m = [complex(x) for x in m]
        is_phase = m[1] == 0 and m[2] == 0
        is_invert = m[0] == 0 and m[3] == 0
        local_c, meta_c = self._split_controls(controls)
        if target < self.qpp:
            # intra-page target; meta controls filter whole pages (semi-meta,
            # zero communication)
            if self._meta_controls_satisfied(meta_c):
                if local_c:
                    self.q.mcmtrx(local_c, m, target)
                else:
                    self.q.mtrx(m, target)
            return
        tb = target - self.qpp
        has_tb_control = tb in meta_c
        if is_phase:
            if self._pair_meta_controls_satisfied(meta_c, tb):
                self._meta_phase(local_c, m[0], m[3], tb, has_tb_control)
        elif is_invert:
            if has_tb_control:
                raise ValueError("gate controlled on its own target qubit")
            # NOTE: must run on EVERY rank — partial page relabeling mutates
            # the replicated page_of_rank table
            self._meta_invert(local_c, meta_c, m[1], m[2], tb)
        else:
            if has_tb_control:
                raise ValueError("gate controlled on its own target qubit")
            if self._pair_meta_controls_satisfied(meta_c, tb):
                self._meta_mtrx(local_c, m, tb)

    def mcphase(self, controls, tl, br, target):
        self.mcmtrx(controls, [tl, 0, 0, br], target)

    def mcinvert(self, controls, tr, bl, target):
        self.mcmtrx(controls, [0, tr, bl, 0], target)

    def _scale_local(self, local_c, factor):
        """Multiply amplitudes with all local controls set by `factor`."""
        if factor == 1:
            return
        if not local_c:
            self.q.global_phase(factor)
        elif len(local_c) == 1:
            self.q.phase(1, factor, local_c[0])
        else:
            self.q.mcphase(local_c[:-1], 1, factor, local_c[-1])

    def _meta_phase(self, local_c, tl, br, tb, has_tb_control):
        bit = (self.my_page >> tb) & 1
        if has_tb_control:
            # control on the target bit itself: only bit=1 pages scale (by br)
            if bit:
                self._scale_local(local_c, br)
            return
        self._scale_local(local_c, br if bit else tl)

    def _meta_invert(self, local_c, meta_c, tr, bl, tb):
        participates = self._pair_meta_controls_satisfied(meta_c, tb)
        if local_c:
            # invert with local controls mixes controlled and uncontrolled
            # amplitudes ACROSS pages: needs the exchange path
            if participates:
                self._meta_mtrx(local_c, [0, tr, bl, 0], tb)
            return
        # partial page relabel (+ per-page scalar) — zero data motion.
        # Pages whose meta controls are satisfied swap labels with their
        # tb-partner; the permutation is computed identically on every rank.
        mask = 1 << tb
        ctrl_bits = [b for b in meta_c if b != tb]

        def sat(p):
            return all((p >> b) & 1 for b in ctrl_bits)

        self.page_of_rank = [p ^ mask if sat(p) else p for p in self.page_of_rank]
        if participates:
            new_bit = (self.my_page >> tb) & 1
            factor = tr if new_bit == 0 else bl
            if factor != 1:
                self.q.global_phase(factor)

    def _meta_mtrx(self, local_c, m, tb):
        # ShuffleBuffers sandwich (reference: qpager.cpp:369-448): after the
        # half-swap, the local top qubit (qpp-1) indexes the meta target bit
        # on BOTH pages of the pair, so the same 2x2 applies per page; the
        # second swap restores the layout.
        partner_page = self.my_page ^ (1 << tb)
        partner_rank = self._rank_of_page(partner_page)
        i_am_low = ((self.my_page >> tb) & 1) == 0
        top = self.qpp - 1
        ctrls = list(local_c)
        top_controlled = top in ctrls
        if top_controlled:
            # in the shuffled layout the ORIGINAL local bit (qpp-1) is
            # page-constant: 0 on the low page, 1 on the high page
            ctrls.remove(top)
        self._shuffle(partner_rank, i_am_low)
        if not (top_controlled and i_am_low):
            if ctrls:
                self.q.mcmtrx(ctrls, m, top)
            else:
                self.q.mtrx(m, top)
        self._shuffle(partner_rank, i_am_low)

    # ---- named gates ---------------------------------------------------------

    def h(self, t):
        s = 1 / np.sqrt(2)
        self.mtrx([s, s, s, -s], t)

    def x(self, t):
        self.mcmtrx([], [0, 1, 1, 0], t)

    def y(self, t):
        self.mcmtrx([], [0, -1j, 1j, 0], t)

    def z(self, t):
        self.phase(1, -1, t)

    def s(self, t):
        self.phase(1, 1j, t)

    def t(self, t_):
        self.phase(1, np.exp(1j * np.pi / 4), t_)

    def rz(self, theta, t):
        self.phase(np.exp(-1j * theta / 2), np.exp(1j * theta / 2), t)

    def cnot(self, c, t):
        self.mcinvert([c], 1, 1, t)

    def cz(self, c, t):
        self.mcphase([c], 1, -1, t)

    def cphase_root_n(self, n, c, t):
        if n == 0:
            return
        self.mcphase([c], 1, np.exp(1j * np.pi / (1 << (n - 1))), t)

    def swap(self, a, b):
        if a == b:
            return
        if a < self.qpp and b < self.qpp:
            self.q.swap(a, b)
            return
        if a >= self.qpp and b >= self.qpp:
            # meta-meta swap: pure page relabel
            ba, bb = a - self.qpp, b - self.qpp
            self.page_of_rank = [
                self._swap_bits(p, ba, bb) for p in self.page_of_rank
            ]
            return
        # local<->meta swap via 3 CNOTs (one comm round for the meta-target one)
        self.cnot(a, b)
        self.cnot(b, a)
        self.cnot(a, b)

    @staticmethod
    def _swap_bits(v, i, j):
        bi, bj = (v >> i) & 1, (v >> j) & 1
        if bi != bj:
            v ^= (1 << i) | (1 << j)
        return v

    # ---- QFT ------------------------------------------------------------------
    # column-fused: the controlled-phase ladder of column i collapses to ONE
    # per-rank PhaseRamp kernel (+ a per-page scalar for meta bits); the only
    # communication is the H exchange on meta columns (mirrors QPager::QFT in
    # csrc/qpager.cpp)

    def _qft_column_ramp(self, start, i, sign):
        t = start + i
        scale = sign * np.pi / (1 << i)
        if t < self.qpp:
            self.q.phase_ramp(scale, start, i, 1 << t)
            return
        tb = t - self.qpp
        if not ((self.my_page >> tb) & 1):
            return
        intra_bits = (self.qpp - start) if start < self.qpp else 0
        meta_start = 0 if start < self.qpp else start - self.qpp
        if intra_bits:
            self.q.phase_ramp(scale, start, intra_bits, 0)
        nbits = tb - meta_start
        meta_val = (self.my_page >> meta_start) & ((1 << nbits) - 1) if nbits > 0 else 0
        if meta_val:
            theta = scale * (meta_val << intra_bits)
            self.q.global_phase(complex(np.exp(1j * theta)))

    def qft(self, start, length):
        for i in range(length - 1, -1, -1):
            self.h(start + i)
            if i:
                self._qft_column_ramp(start, i, +1)

    def iqft(self, start, length):
        for i in range(length):
            if i:
                self._qft_column_ramp(start, i, -1)
            self.h(start + i)

    # ---- measurement ----------------------------------------------------------

    def _local_norm(self):
        return float(self.q.norm_total())

    def _allreduce_scalar(self, v):
        t = torch.tensor([v], dtype=torch.float64)
        dist.all_reduce(t)
        return float(t.item())

    def prob(self, q):
        if q < self.qpp:
            # engine prob is computed on the unnormalized page; it IS this
            # page's contribution to the global probability
            local = float(self.q.prob(q))
            return min(1.0, self._allreduce_scalar(local))
        bit = (self.my_page >> (q - self.qpp)) & 1
        local = self._local_norm() if bit else 0.0
        return min(1.0, self._allreduce_scalar(local))

    def force_m(self, q, result, do_force=True, do_apply=True):
        p1 = self.prob(q)
        if not do_force:
            result = bool(self.rng.random() < p1)  # replicated RNG: same draw
        if do_apply:
            prob = p1 if result else 1.0 - p1
            if prob <= 0:
                raise RuntimeError("impossible measurement outcome")
            nrm = 1.0 / np.sqrt(prob)
            if q < self.qpp:
                self.q.apply_m(1 << q, (1 << q) if result else 0, complex(nrm))
            else:
                bit = (self.my_page >> (q - self.qpp)) & 1
                if bool(bit) == bool(result):
                    self.q.global_phase(complex(nrm))
                else:
                    self.q.zero_amplitudes()
        return result

    def m(self, q):
        return self.force_m(q, False, do_force=False)

    def m_all(self):
        norms = [0.0] * self.world
        t = torch.zeros(self.world, dtype=torch.float64)
        t[self.rank] = self._local_norm()
        dist.all_reduce(t)
        norms = t.tolist()
        total = sum(norms)
        r = self.rng.random() * total  # replicated draw
        owner = 0
        for rank in range(self.world):
            if r <= norms[rank] or rank == self.world - 1:
                owner = rank
                break
            r -= norms[rank]
        res_t = torch.zeros(1, dtype=torch.int64)
        if self.rank == owner:
            res = self.q.multi_shot_measure_mask(
                [1 << i for i in range(self.qpp)], 1
            )
            local_idx = next(iter(res.keys()))
            res_t[0] = (self.page_of_rank[owner] << self.qpp) | local_idx
        dist.broadcast(res_t, src=owner)
        result = int(res_t.item())
        self.set_permutation(result)
        return result

    def multi_shot_measure_mask(self, q_powers, shots):
        t = torch.zeros(self.world, dtype=torch.float64)
        t[self.rank] = self._local_norm()
        dist.all_reduce(t)
        norms = np.maximum(t.numpy(), 0.0)
        total = norms.sum()
        counts = self.rng.multinomial(shots, norms / total)  # replicated
        my_count = int(counts[self.rank])
        local_results = {}
        if my_count > 0:
            # sample local indices, then map through the global page index
            local_powers = [1 << i for i in range(self.qpp)]
            res = self.q.multi_shot_measure_mask(local_powers, my_count)
            page_high = self.my_page << self.qpp
            for local_idx, c in res.items():
                g = page_high | local_idx
                val = 0
                for b, p in enumerate(q_powers):
                    if g & p:
                        val |= 1 << b
                local_results[val] = local_results.get(val, 0) + c
        gathered = [None] * self.world
        dist.all_gather_object(gathered, local_results)
        merged = {}
        for d in gathered:
            for k, v in d.items():
                merged[k] = merged.get(k, 0) + v
        return merged
