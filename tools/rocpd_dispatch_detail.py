"""Aggregate kernel times from a rocprofv3 results DB."""
import sqlite3, re, collections, sys, glob

db = sys.argv[1] if len(sys.argv) > 1 else glob.glob('gpurun_out/prof/*results.db')[0]
con = sqlite3.connect(db)
sfx = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'") if r[0].startswith('rocpd_kernel_dispatch')][0].replace('rocpd_kernel_dispatch_','')
rows = con.execute(f"""
SELECT ks.display_name, kd.start, kd.end, kd.grid_size_x, kd.grid_size_y,
       ks.arch_vgpr_count, ks.accum_vgpr_count, kd.workgroup_size_x
FROM rocpd_kernel_dispatch_{sfx} kd
JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
""").fetchall()
agg = collections.defaultdict(lambda: [0,0.0,0])
for name, st, en, gx, gy, vg, ag, wx in rows:
    m = re.search(r'(conv_igemm_kernel|pad_channels|bn_act|add_act|relu_kernel|softmax|maxpool|gap_kernel)', name)
    key = m.group(1) if m else name[:46]
    if m and m.group(1)=='conv_igemm_kernel':
        key = f"igemm(g={gx//wx}x{gy},v={vg}+{ag})"
    agg[key][0]+=1
    agg[key][1]+=(en-st)/1e6
tot = sum(v[1] for v in agg.values())
nfwd = 13
print(f"total kernel ms: {tot:.1f} -> {tot/nfwd:.2f} ms/fwd (assuming {nfwd} fwds)")
for k,(n,t,_) in sorted(agg.items(), key=lambda kv:-kv[1][1])[:25]:
    print(f"{k:38s} n={n:5d}  {t:8.2f} ms {100*t/tot:5.1f}%  {t/n*1000:7.1f} us/call")
