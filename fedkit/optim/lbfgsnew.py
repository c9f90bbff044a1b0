"""Stochastic L-BFGS with line searches — behavior parity with the reference.

Algorithm parity with reference src/lbfgsnew.py (LBFGSNew):
  * two-loop recursion with history gated by ys > 1e-10 ||s||^2
    (lbfgsnew.py:618-630) and H0 = ys / y.y (656);
  * batch (stochastic) mode adds a trust-region damping y += 1e-6 s
    (594-595) and an online inter-batch gradient mean/variance estimate
    (Welford) that caps the step: alphabar = 1/(1 + var_sum/((n-1)||g||))
    (600-613);
  * full-batch line search: strong-Wolfe bracketing with cubic
    interpolation and zoom, derivatives by central differencing of the
    closure (201-325, 328-504);
  * batch-mode line search: Armijo backtracking with halving and a
    negative-step rescue (124-196);
  * same constructor signature and defaults, one parameter group only.

MI355X design: all vector state (gradients, direction, history pairs) lives
in flat fp32 HBM buffers; gather/scatter/axpy over the parameter list run as
single fused multi-tensor HIP kernels (fedkit.ops.flat) instead of the
reference's per-tensor Python loops.  Line-search probes are loss-only
closure evaluations (grad disabled), cheap enough to keep on-stream.
"""

import math
from functools import reduce

import torch
from torch.optim.optimizer import Optimizer

from ..ops import flat as flat_ops

# Armijo backtracking constants (reference lbfgsnew.py:137-140)
_ARMIJO_C1 = 1e-4
_ARMIJO_MAX_HALVINGS = 35
# Strong-Wolfe / Fletcher constants (reference lbfgsnew.py:211-217)
_WOLFE_SIGMA = 0.1
_WOLFE_RHO = 0.01
_WOLFE_T1 = 9.0
_WOLFE_T2 = 0.1
_WOLFE_T3 = 0.5
# trust-region damping coefficient in batch mode (lbfgsnew.py:559)
_BATCH_LM0 = 1e-6
# Max vectors per fused multi_dot call (csrc/flat_ops.hip kMaxVecs).
_FUSED_MAX_VECS = 24


class LBFGSNew(Optimizer):
    """L-BFGS with cubic/backtracking line search and a stochastic mode.

    Args:
        params: iterable of parameters (one group, one device).
        lr: fallback step size when line search is off / fails.
        max_iter: L-BFGS iterations per .step(closure).
        max_eval: max closure evaluations per step (default max_iter*5//4).
        tolerance_grad / tolerance_change: termination tolerances.
        history_size: curvature-pair memory.
        line_search_fn: True => cubic strong-Wolfe (full batch) or Armijo
            backtracking (batch_mode); False => fixed step lr.
        batch_mode: True for the stochastic variant.
    """

    def __init__(self, params, lr=1, max_iter=10, max_eval=None,
                 tolerance_grad=1e-5, tolerance_change=1e-9, history_size=7,
                 line_search_fn=False, batch_mode=False):
        if max_eval is None:
            max_eval = max_iter * 5 // 4
        defaults = dict(lr=lr, max_iter=max_iter, max_eval=max_eval,
                        tolerance_grad=tolerance_grad,
                        tolerance_change=tolerance_change,
                        history_size=history_size,
                        line_search_fn=line_search_fn, batch_mode=batch_mode)
        super().__init__(params, defaults)
        if len(self.param_groups) != 1:
            raise ValueError("LBFGSNew supports a single parameter group only")
        self._params = self.param_groups[0]["params"]
        self._numel_cache = None

    # ----------------------------------------------------------- flat plumbing

    def _numel(self):
        if self._numel_cache is None:
            self._numel_cache = reduce(lambda acc, p: acc + p.numel(), self._params, 0)
        return self._numel_cache

    def _flat_grad(self):
        # layouts preserved: the pack kernel traverses PHYSICAL storage
        # order, matching _move_along's add_flat over p.data — a standard
        # .contiguous() here would reorder channels_last conv-weight grads
        # and permute the applied direction
        grads = []
        for p in self._params:
            if p.grad is None:
                grads.append(torch.zeros_like(p.data))
            elif p.grad.is_sparse:
                grads.append(p.grad.to_dense())
            else:
                grads.append(p.grad)
        return flat_ops.pack([g.data.float() for g in grads])

    def _move_along(self, alpha, direction):
        """params += alpha * direction (flat)."""
        flat_ops.add_flat([p.detach() for p in self._params], direction, alpha)

    def _snapshot(self):
        return flat_ops.pack([p.detach() for p in self._params])

    def _restore(self, snap):
        flat_ops.unpack(snap, [p.detach() for p in self._params])

    # ----------------------------------------------------------- line searches

    def _armijo_backtrack(self, closure, pk, gk, alphabar):
        """Armijo backtracking with negative-step rescue (lbfgsnew.py:124-196)."""
        state = self.state[self._params[0]]
        alphak = alphabar
        snap = self._snapshot()

        f_old = float(closure())
        self._move_along(alphak, pk)
        f_new = float(closure())
        prodterm = _ARMIJO_C1 * float(gk.dot(pk))

        halvings = 0
        while halvings < _ARMIJO_MAX_HALVINGS and (
                math.isnan(f_new) or f_new > f_old + alphak * prodterm):
            alphak *= 0.5
            self._restore(snap)
            self._move_along(alphak, pk)
            f_new = float(closure())
            halvings += 1

        # insufficient decrease: probe the negative direction too
        if f_old - f_new < abs(prodterm):
            alphak_neg = -alphabar
            self._restore(snap)
            self._move_along(alphak_neg, pk)
            f_neg = float(closure())
            while halvings < _ARMIJO_MAX_HALVINGS and (
                    math.isnan(f_neg) or f_neg > f_old + alphak_neg * prodterm):
                alphak_neg *= 0.5
                self._restore(snap)
                self._move_along(alphak_neg, pk)
                f_neg = float(closure())
                halvings += 1
            if f_neg < f_new:
                alphak = alphak_neg

        self._restore(snap)
        state["func_evals"] += halvings
        return alphak

    def _numeric_slope(self, closure, pk, step):
        """Central-difference directional derivative at the CURRENT params.

        Leaves params displaced by -step*pk relative to entry (caller
        compensates), matching the reference's probe bookkeeping.
        """
        self._move_along(step, pk)
        f_plus = float(closure())
        self._move_along(-2.0 * step, pk)
        f_minus = float(closure())
        return (f_plus - f_minus) / (2.0 * step)

    def _wolfe_cubic(self, closure, pk, step):
        """Strong-Wolfe bracketing line search (lbfgsnew.py:201-325)."""
        lr = self.param_groups[0]["lr"]
        alpha1 = 10.0 * lr
        alphak = lr
        state = self.state[self._params[0]]
        snap = self._snapshot()

        phi_0 = float(closure())
        tol = min(phi_0 * 0.01, 1e-6)

        # numeric phi'(0)
        self._move_along(step, pk)
        p_plus = float(closure())
        self._move_along(-2.0 * step, pk)
        p_minus = float(closure())
        gphi_0 = (p_plus - p_minus) / (2.0 * step)
        if abs(gphi_0) < 1e-12:
            self._restore(snap)
            return 1.0
        mu = (tol - phi_0) / (_WOLFE_RHO * gphi_0)
        if math.isnan(mu):
            self._restore(snap)
            return 1.0

        evals = 3
        it = 1
        alphai = alpha1
        alphai_prev = 0.0
        phi_prev = phi_0
        while it < 4:
            self._restore(snap)
            self._move_along(alphai, pk)
            phi_i = float(closure())
            if phi_i < tol:
                alphak = alphai
                break
            if (phi_i > phi_0 + alphai * gphi_0) or (it > 1 and phi_i >= phi_prev):
                alphak = self._zoom(closure, snap, pk, alphai_prev, alphai,
                                    phi_0, gphi_0, step)
                break
            # numeric phi'(alphai); params currently at x + alphai*pk
            gphi_i = self._numeric_slope(closure, pk, step)
            if abs(gphi_i) <= -_WOLFE_SIGMA * gphi_0:
                alphak = alphai
                break
            if gphi_i >= 0.0:
                alphak = self._zoom(closure, snap, pk, alphai, alphai_prev,
                                    phi_0, gphi_0, step)
                break
            if mu <= 2.0 * alphai - alphai_prev:
                alphai_prev, alphai = alphai, mu
            else:
                lo = 2.0 * alphai - alphai_prev
                hi = min(mu, alphai + _WOLFE_T1 * (alphai - alphai_prev))
                new_alpha = self._cubic_min(closure, snap, pk, lo, hi, step)
                alphai_prev, alphai = alphai, new_alpha
            phi_prev = phi_i
            evals += 3
            it += 1

        self._restore(snap)
        state["func_evals"] += evals
        return alphak

    def _cubic_min(self, closure, snap, pk, a, b, step):
        """Cubic interpolation on [a,b] (order free; lbfgsnew.py:328-418)."""
        state = self.state[self._params[0]]
        self._restore(snap)

        self._move_along(a, pk)
        f0 = float(closure())
        f0d = self._numeric_slope(closure, pk, step)
        # params now at a - step; move to b
        self._move_along(-a + step + b, pk)
        f1 = float(closure())
        f1d = self._numeric_slope(closure, pk, step)
        evals = 6

        aa = 3.0 * (f0 - f1) / (b - a) + f1d - f0d
        disc = aa * aa - f0d * f1d
        if disc > 0.0:
            cc = math.sqrt(disc)
            denom = f1d - f0d + 2.0 * cc
            if denom == 0.0:
                return (a + b) * 0.5
            z0 = b - (f1d + cc - aa) * (b - a) / denom
            hi, lo = max(a, b), min(a, b)
            if z0 > hi or z0 < lo:
                fz0 = f0 + f1
            else:
                # params sit at b - step; move to a + z0*(b-a)
                self._move_along(-b + step + a + z0 * (b - a), pk)
                fz0 = float(closure())
                evals += 1
            state["func_evals"] += evals
            if f0 < f1 and f0 < fz0:
                return a
            if f1 < fz0:
                return b
            return z0
        state["func_evals"] += evals
        return a if f0 < f1 else b

    def _zoom(self, closure, snap, pk, a, b, phi_0, gphi_0, step):
        """Zoom phase of the Wolfe search (lbfgsnew.py:421-504)."""
        state = self.state[self._params[0]]
        evals = 0
        aj, bj = a, b
        alphaj = aj
        for _ in range(4):
            lo = aj + _WOLFE_T2 * (bj - aj)
            hi = bj - _WOLFE_T3 * (bj - aj)
            alphaj = self._cubic_min(closure, snap, pk, lo, hi, step)

            self._restore(snap)
            self._move_along(alphaj, pk)
            phi_j = float(closure())
            self._move_along(-alphaj + aj, pk)
            phi_aj = float(closure())
            evals += 2

            if (phi_j > phi_0 + _WOLFE_RHO * alphaj * gphi_0) or phi_j >= phi_aj:
                bj = alphaj
            else:
                # numeric phi'(alphaj); params currently at x + aj*pk
                self._move_along(-aj + alphaj, pk)
                gphi_j = self._numeric_slope(closure, pk, step)
                evals += 2
                if (aj - alphaj) * gphi_j <= step:      # roundoff guard (Fletcher p.38)
                    break
                if abs(gphi_j) <= -_WOLFE_SIGMA * gphi_0:
                    break
                if gphi_j * (bj - aj) >= 0.0:
                    bj = aj
                aj = alphaj
        state["func_evals"] += evals
        return alphaj

    # ------------------------------------------------------------------- step

    @torch.no_grad()
    def _two_loop(self, flat_grad, hist_y, hist_s, H_diag, ro, al):
        q = flat_grad.neg()
        n = len(hist_y)
        for i in range(n):
            ro[i] = 1.0 / float(hist_y[i].dot(hist_s[i]))
        for i in range(n - 1, -1, -1):
            al[i] = float(hist_s[i].dot(q)) * ro[i]
            q.add_(hist_y[i], alpha=-al[i])
        r = torch.mul(q, H_diag)
        for i in range(n):
            be_i = float(hist_y[i].dot(r)) * ro[i]
            r.add_(hist_s[i], alpha=al[i] - be_i)
        return r

    @torch.no_grad()
    def _two_loop_fused(self, flat_grad, hist_y, hist_s, H_diag, SY, YY):
        """Two-loop recursion via the compact (Gram-matrix) representation.

        q and r stay in span{g, y_*, s_*}, so every s_i.q / y_i.r the
        reference computes on the full N-vector (lbfgsnew.py:645-659)
        reduces to algebra over SY[i][j] = s_i.y_j and YY[i][j] = y_i.y_j
        (maintained incrementally on history updates) plus the 2n fresh
        dots (s_i.g, y_i.g) from ONE fused multi_dot pass.  The direction
        is then materialized by one fused lincomb kernel: ~4 launches and
        one device sync per iteration instead of ~3n launches and ~2n syncs.
        """
        n = len(hist_y)
        if n == 0:
            return flat_ops.lincomb(flat_grad, -float(H_diag), [], [])
        dots = flat_ops.multi_dot(hist_s + hist_y, flat_grad).tolist()
        sg, yg = dots[:n], dots[n:]
        ro = [1.0 / SY[i][i] for i in range(n)]
        # backward pass: q = qg*g + sum qy[j]*y_j
        qg, qy = -1.0, [0.0] * n
        al = [0.0] * n
        for i in range(n - 1, -1, -1):
            siq = qg * sg[i] + sum(qy[j] * SY[i][j] for j in range(n))
            al[i] = siq * ro[i]
            qy[i] -= al[i]
        # forward pass: r = rg*g + sum ry[j]*y_j + sum rs[j]*s_j
        rg = H_diag * qg
        ry = [H_diag * c for c in qy]
        rs = [0.0] * n
        for i in range(n):
            yir = rg * yg[i] + sum(ry[j] * YY[i][j] for j in range(n)) \
                + sum(rs[j] * SY[j][i] for j in range(n))
            rs[i] += al[i] - yir * ro[i]
        return flat_ops.lincomb(flat_grad, rg, hist_y + hist_s, ry + rs)

    def step(self, closure):
        assert len(self.param_groups) == 1
        group = self.param_groups[0]
        lr = group["lr"]
        max_iter = group["max_iter"]
        max_eval = group["max_eval"]
        tolerance_grad = group["tolerance_grad"]
        tolerance_change = group["tolerance_change"]
        line_search_fn = group["line_search_fn"]
        history_size = group["history_size"]
        batch_mode = group["batch_mode"]

        # global state registered on the first parameter (load_state_dict compat)
        state = self.state[self._params[0]]
        state.setdefault("func_evals", 0)
        state.setdefault("n_iter", 0)

        orig_loss = closure()
        loss = float(orig_loss.detach() if torch.is_tensor(orig_loss)
                     else orig_loss)
        current_evals = 1
        state["func_evals"] += 1

        flat_grad = self._flat_grad()
        abs_grad_sum = float(flat_grad.abs().sum())
        if abs_grad_sum <= tolerance_grad:
            return orig_loss

        d = state.get("d")
        t = state.get("t")
        hist_y = state.get("old_dirs")
        hist_s = state.get("old_stps")
        H_diag = state.get("H_diag")
        prev_flat_grad = state.get("prev_flat_grad")
        prev_loss = state.get("prev_loss")

        if batch_mode:
            alphabar = lr
        grad_nrm = float(flat_grad.norm())

        n_iter = 0
        while n_iter < max_iter and not math.isnan(grad_nrm):
            n_iter += 1
            state["n_iter"] += 1

            # ---- direction
            if state["n_iter"] == 1:
                d = flat_grad.neg()
                hist_y, hist_s = [], []
                state["SY"], state["YY"] = [], []
                H_diag = 1
                if batch_mode:
                    running_avg = torch.zeros_like(flat_grad)
                    running_avg_sq = torch.zeros_like(flat_grad)
            else:
                if batch_mode:
                    running_avg = state.get("running_avg")
                    running_avg_sq = state.get("running_avg_sq")
                    if running_avg is None:
                        running_avg = torch.zeros_like(flat_grad)
                        running_avg_sq = torch.zeros_like(flat_grad)

                y = flat_grad.sub(prev_flat_grad)
                s = d.mul(t)
                if batch_mode:
                    y.add_(s, alpha=_BATCH_LM0)      # trust-region damping

                # fused path: every scalar the update and the two-loop need
                # comes from two multi_dot passes and ONE host sync.
                # multi_dot takes at most _FUSED_MAX_VECS vectors per call
                # (csrc/flat_ops.hip kMaxVecs); the largest call below is
                # hist_s + hist_y + [s, y] = 2*history+2, so history sizes
                # past (kMaxVecs-2)//2 fall back to the unfused two-loop
                # instead of aborting mid-training.
                fused = flat_ops.fused_available(flat_grad) \
                    and history_size <= (_FUSED_MAX_VECS - 2) // 2
                SY, YY = state.get("SY", []), state.get("YY", [])
                n0 = len(hist_y)
                if fused:
                    d1 = flat_ops.multi_dot(hist_s + hist_y + [s, y], y)
                    d2 = flat_ops.multi_dot(hist_y + [s], s)
                    vals = torch.cat([d1, d2]).tolist()
                    s_y = vals[:n0]                  # s_i . y_new
                    y_y = vals[n0:2 * n0]            # y_i . y_new
                    ys = vals[2 * n0]
                    yy = vals[2 * n0 + 1]
                    y_s = vals[2 * n0 + 2:3 * n0 + 2]  # y_j . s_new
                    sn2 = vals[3 * n0 + 2]
                else:
                    ys = float(y.dot(s))
                    sn2 = float(s.norm()) ** 2

                # first iteration of a new step() call == new minibatch
                batch_changed = batch_mode and (n_iter == 1 and state["n_iter"] > 1)
                if batch_changed:
                    # online inter-batch Welford estimate drives alphabar;
                    # fused path: ONE kernel updates avg/avg_sq in place
                    # and returns sum(avg_sq) (csrc welford_update) vs the
                    # ~6-launch clone/axpy/addcmul/sum chain
                    if fused:
                        from ..ops import require_ext
                        sq_sum = float(require_ext().welford_update(
                            flat_grad, running_avg, running_avg_sq,
                            1.0 / state["n_iter"]))
                    else:
                        g_del_old = flat_grad.clone().add_(running_avg, alpha=-1.0)
                        running_avg.add_(g_del_old, alpha=1.0 / state["n_iter"])
                        g_del_new = flat_grad.clone().add_(running_avg, alpha=-1.0)
                        running_avg_sq.addcmul_(g_del_new, g_del_old, value=1)
                        sq_sum = float(running_avg_sq.sum())
                    alphabar = 1.0 / (1.0 + sq_sum
                                      / ((state["n_iter"] - 1) * grad_nrm))

                if ys > 1e-10 * sn2 and not batch_changed:
                    if len(hist_y) == history_size:
                        hist_y.pop(0)
                        hist_s.pop(0)
                        if fused:
                            SY.pop(0)
                            YY.pop(0)
                            for row in SY:
                                row.pop(0)
                            for row in YY:
                                row.pop(0)
                            s_y, y_y, y_s = s_y[1:], y_y[1:], y_s[1:]
                    hist_y.append(y)
                    hist_s.append(s)
                    if fused:
                        for i, row in enumerate(SY):
                            row.append(s_y[i])
                        SY.append(list(y_s) + [ys])
                        for i, row in enumerate(YY):
                            row.append(y_y[i])
                        YY.append(list(y_y) + [yy])
                        H_diag = ys / yy
                    else:
                        H_diag = ys / float(y.dot(y))

                if isinstance(H_diag, float) and math.isnan(H_diag):
                    print("Warning H_diag nan")

                if fused:
                    d = self._two_loop_fused(flat_grad, hist_y, hist_s,
                                             H_diag, SY, YY)
                else:
                    if "ro" not in state:
                        state["ro"] = [None] * history_size
                        state["al"] = [None] * history_size
                    d = self._two_loop(flat_grad, hist_y, hist_s, H_diag,
                                       state["ro"], state["al"])

            if prev_flat_grad is None:
                prev_flat_grad = flat_grad.clone()
            else:
                prev_flat_grad.copy_(flat_grad)
            prev_loss = loss

            # ---- step length
            if state["n_iter"] == 1:
                t = min(1.0, 1.0 / abs_grad_sum) * lr
            else:
                t = lr

            gtd = float(flat_grad.dot(d))
            if math.isnan(gtd):
                print("Warning grad norm infinite (iter %d)" % state["n_iter"])

            ls_func_evals = 0
            if line_search_fn:
                with torch.set_grad_enabled(False):
                    if not batch_mode:
                        t = self._wolfe_cubic(closure, d, 1e-6)
                    else:
                        t = self._armijo_backtrack(closure, d, flat_grad, alphabar)
                if math.isnan(t):
                    print("Warning: stepsize nan")
                    t = lr
                self._move_along(t, d)
            else:
                self._move_along(t, d)

            if n_iter != max_iter:
                # re-evaluate (new grads for the next inner iteration)
                _l = closure()
                loss = float(_l.detach() if torch.is_tensor(_l) else _l)
                flat_grad = self._flat_grad()
                abs_grad_sum = float(flat_grad.abs().sum())
                if math.isnan(abs_grad_sum):
                    print("Warning: gradient nan")
                    break
                ls_func_evals = 1

            current_evals += ls_func_evals
            state["func_evals"] += ls_func_evals

            # ---- convergence checks
            if n_iter == max_iter:
                break
            if current_evals >= max_eval:
                break
            if abs_grad_sum <= tolerance_grad:
                break
            if gtd > -tolerance_change:
                break
            if float(d.mul(t).abs().sum()) <= tolerance_change:
                break
            if abs(loss - prev_loss) < tolerance_change:
                break

        state["d"] = d
        state["t"] = t
        state["old_dirs"] = hist_y
        state["old_stps"] = hist_s
        state["H_diag"] = H_diag
        state["prev_flat_grad"] = prev_flat_grad
        state["prev_loss"] = prev_loss
        if batch_mode:
            if "running_avg" not in locals() or running_avg is None:  # pragma: no cover
                running_avg = torch.zeros_like(flat_grad)
                running_avg_sq = torch.zeros_like(flat_grad)
            state["running_avg"] = running_avg
            state["running_avg_sq"] = running_avg_sq
        return orig_loss
