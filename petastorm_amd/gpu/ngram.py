"""Batched NGram windowing over HBM-resident decoded columns.

The reference assembles windows with a per-row Python loop
(reference petastorm/ngram.py:225-270 form_ngram).  On MI355X the decoded
row-group lives in HBM as column tensors, so the window rule becomes three
tensor ops (the ``ngram_window_gather`` design of SURVEY.md §2.4 — expressed
through torch's ROCm kernels since it is pure gather/compare, with no custom
bit-twiddling to justify hand-written HIP):

1. sort rows by the timestamp column
2. a sliding-window delta check finds the valid window starts
3. ``index_select`` gathers each timestep's rows

Validated index-for-index against :meth:`NGram.form_ngram_indices` (the CPU
source of truth) in tests/test_gpu_ngram.py.
"""

import torch


def window_starts(timestamps_sorted, length, delta_threshold,
                  timestamp_overlap=True):
    """Valid window start indices over an already-sorted timestamp tensor.

    Mirrors NGram._window_passes_threshold (reference ngram.py:179-193): a
    start s is valid when every consecutive delta within
    ``[s, s+length)`` is <= delta_threshold.
    """
    n = timestamps_sorted.numel()
    if n < length:
        return torch.empty(0, dtype=torch.int64,
                           device=timestamps_sorted.device)
    if length == 1:
        starts = torch.arange(n, device=timestamps_sorted.device)
    else:
        deltas = timestamps_sorted[1:] - timestamps_sorted[:-1]
        ok = (deltas <= delta_threshold)
        # window s valid iff ok[s : s+length-1] all true
        okf = ok.to(torch.float32)
        window = torch.ones(length - 1, dtype=torch.float32,
                            device=ok.device)
        conv = torch.nn.functional.conv1d(
            okf.view(1, 1, -1), window.view(1, 1, -1)).view(-1)
        starts = torch.nonzero(conv >= (length - 1) - 0.5,
                               as_tuple=False).squeeze(1)
    if not timestamp_overlap and starts.numel():
        # greedy non-overlap selection is order-serial; do it host-side on
        # the (small) starts list (reference ngram.py:107-117)
        s_cpu = starts.cpu().tolist()
        keep = []
        nxt = 0
        for s in s_cpu:
            if s >= nxt:
                keep.append(s)
                nxt = s + length
        starts = torch.tensor(keep, dtype=torch.int64,
                              device=timestamps_sorted.device)
    return starts


def form_ngram_batched(columns, ngram, timestamp_column=None):
    """Assemble all windows of one decoded row-group.

    :param columns: dict name -> tensor [n_rows, ...] (device or cpu)
    :param ngram: :class:`petastorm_amd.ngram.NGram`
    :return: dict ``{timestep: {field_name: tensor[n_windows, ...]}}``
    """
    ts_name = timestamp_column or ngram.timestamp_field_name
    ts = columns[ts_name]
    order = torch.argsort(ts, stable=True)
    starts = window_starts(ts.index_select(0, order), ngram.length,
                           ngram.delta_threshold, ngram.timestamp_overlap)
    base = min(ngram.fields.keys())
    out = {}
    for offset_idx in range(ngram.length):
        ts_key = base + offset_idx
        idx = order.index_select(0, starts + offset_idx)
        names = ngram.get_field_names_at_timestep(ts_key)
        out[ts_key] = {name: columns[name].index_select(0, idx)
                       for name in names if name in columns}
    return out
