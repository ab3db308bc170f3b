"""Volume-profile analytics (reference parity:
services/utils/volume_profile_analyzer.py — price-binned volume histogram,
POC + 70% value area, POC/VA signal rules, buy/sell delta, anomaly
z-scores). numpy implementation plus the GPU histogram-kernel path
(vp_hist_gpu -> ops/hip/histogram.hip) for bulk histories."""

from __future__ import annotations

import numpy as np


class VolumeProfileAnalyzer:
    def __init__(self, n_bins: int = 24, value_area_pct: float = 0.70):
        self.n_bins = n_bins
        self.value_area_pct = value_area_pct

    def analyze(self, candles: np.ndarray) -> dict:
        """candles: (T, 4) [close, high, low, volume] -> profile dict
        (volume_profile_analyzer.py:86-174 semantics)."""
        close = candles[:, 0]
        vol = candles[:, 3]
        lo, hi = float(candles[:, 2].min()), float(candles[:, 1].max())
        if hi <= lo:
            hi = lo * 1.0001 + 1e-9
        hist, edges = np.histogram(
            close, bins=self.n_bins, range=(lo, hi), weights=vol)
        centers = 0.5 * (edges[:-1] + edges[1:])
        poc_i = int(np.argmax(hist))
        poc = float(centers[poc_i])

        # value area: expand around POC until >= value_area_pct of volume
        total = float(hist.sum()) or 1.0
        inc = {poc_i}
        acc = hist[poc_i]
        l, r = poc_i - 1, poc_i + 1
        while acc / total < self.value_area_pct and (l >= 0 or r < len(hist)):
            lv = hist[l] if l >= 0 else -1
            rv = hist[r] if r < len(hist) else -1
            if lv >= rv:
                inc.add(l); acc += max(lv, 0); l -= 1
            else:
                inc.add(r); acc += max(rv, 0); r += 1
        va_lo = float(edges[min(inc)])
        va_hi = float(edges[max(inc) + 1])

        # buy/sell volume delta (:564-686): up-candle volume vs down-candle
        up = vol[1:][close[1:] >= close[:-1]].sum()
        dn = vol[1:][close[1:] < close[:-1]].sum()
        delta = float((up - dn) / max(up + dn, 1e-9))

        # anomaly z-score of the most recent volume (:487)
        mu, sd = float(vol.mean()), float(vol.std() + 1e-9)
        z_last = float((vol[-1] - mu) / sd)

        price = float(close[-1])
        if price > va_hi:
            signal = "above_value_area"       # (:232-317 signal rules)
        elif price < va_lo:
            signal = "below_value_area"
        elif abs(price - poc) / max(poc, 1e-9) < 0.001:
            signal = "at_poc"
        else:
            signal = "inside_value_area"

        return {
            "poc": poc,
            "value_area_low": va_lo,
            "value_area_high": va_hi,
            "volume_delta": delta,
            "volume_zscore_last": z_last,
            "signal": signal,
            "bins": [float(x) for x in hist],
            "bin_edges": [float(x) for x in edges],
        }


def vp_hist_gpu(candles, n_bins: int = 24):
    """Batched GPU volume histograms: (nsym, T, 4) f32 cuda ->
    (hist (nsym, n_bins), updown (nsym, 2), lo, hi). Kernel:
    ops/hip/histogram.hip (LDS-atomic privatized histograms)."""
    import torch

    from ..ops import require_hip_ops

    ops = require_hip_ops()
    assert candles.is_cuda and candles.dtype == torch.float32
    candles = candles.contiguous()
    nsym, T, _ = candles.shape
    lo = candles[:, :, 2].amin(dim=1).contiguous()
    hi = (candles[:, :, 1].amax(dim=1) + 1e-9).contiguous()
    hist = torch.empty(nsym, n_bins, dtype=torch.float32,
                       device=candles.device)
    updown = torch.empty(nsym, 2, dtype=torch.float32,
                         device=candles.device)
    stream = torch.cuda.current_stream(candles.device).cuda_stream
    ops.vp_hist(candles.data_ptr(), lo.data_ptr(), hi.data_ptr(),
                hist.data_ptr(), updown.data_ptr(), nsym, T, n_bins,
                stream)
    return hist, updown, lo, hi
