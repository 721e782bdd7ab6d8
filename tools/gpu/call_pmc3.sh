cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_INSTS_VALU TCC_HIT TCC_MISS SQ_WAVES --kernel-include-regex "jpeg_color_norm" -d gpurun_out/pmc_a -- python bench.py --config imagenet --steps 3 --warmup 2 --min-region 1 > /dev/null 2>&1
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_INSTS_VALU TCC_HIT TCC_MISS SQ_WAVES --kernel-include-regex "snappy_decompress" -d gpurun_out/pmc_b -- python bench.py --config scalar --steps 3 --warmup 2 --min-region 1 > /dev/null 2>&1
python - > gpurun_out/pmc_summary_r2.txt 2>&1 <<'PY'
import sqlite3, glob
for tag, pat in (('jpeg_color_norm', 'gpurun_out/pmc_a/runc/*_results.db'),
                 ('snappy_decompress (scalar config)', 'gpurun_out/pmc_b/runc/*_results.db')):
    try:
        db = sqlite3.connect(glob.glob(pat)[0])
        cur = db.cursor()
        t = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'") if 'pmc_event' in r[0] and r[0].startswith('rocpd_pmc_event')][0]
        sfx = t[len('rocpd_pmc_event'):]
        rows = dict((n, v) for n, v in cur.execute(
            "SELECT i.name, SUM(e.value) FROM %s e JOIN rocpd_info_pmc%s i ON e.pmc_id=i.id GROUP BY i.name" % (t, sfx)))
        print('== %s ==' % tag)
        for k in sorted(rows):
            print('  %-16s %18.0f' % (k, rows[k]))
        wc, wa = rows.get('SQ_WAVE_CYCLES',0), rows.get('SQ_WAIT_ANY',0)
        th, tm = rows.get('TCC_HIT',0), rows.get('TCC_MISS',0)
        if wc: print('  stall-frac %.3f  valu-frac %.3f' % (wa/wc, rows.get('SQ_INSTS_VALU',0)/wc))
        if th+tm: print('  L2 hit %.3f' % (th/(th+tm)))
    except Exception as e:
        print(tag, 'FAILED:', e)
PY
rm -rf gpurun_out/pmc_a gpurun_out/pmc_b
cat gpurun_out/pmc_summary_r2.txt
