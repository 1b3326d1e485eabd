import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, oracle
from neutronstarlite_amd import graph as G
from neutronstarlite_amd.ops import DeviceChunk, HipEngine, SingleGPUFuseOp
dev = torch.device("cuda:0")
for f in [1, 128, 602]:
    v, e = 3000, 60000
    edges = G.rmat_edges(v, e, seed=7)
    outd, ind = G.degrees(edges, v)
    w = G.norm_weights(edges[:,0], edges[:,1], outd, ind)
    ch = G.build_chunks(edges, w, np.array([0,v],np.uint32), 0)[0]
    rng = np.random.default_rng(42)
    x = rng.uniform(-1,1,(v,f)).astype(np.float32)
    op = SingleGPUFuseOp(DeviceChunk(ch, dev), HipEngine())
    xt = torch.from_numpy(x).to(dev)
    y1 = op.forward(xt); torch.cuda.synchronize()
    y2 = op.forward(xt); torch.cuda.synchronize()
    det = torch.equal(y1, y2)
    y = y1.cpu().numpy()
    ref = oracle.csc_forward(ch.column_offset, ch.row_indices, ch.edge_weight_forward, x, 0, v, f)
    err = np.abs(y-ref); tol = 1e-4*np.abs(ref)+1e-5
    bad = err > tol
    nb = int(bad.sum())
    print(f"f={f}: det={det} bad={nb}/{bad.size} worst={float(err.max()):.3e}")
    if nb:
        r, c = np.where(bad)
        print("  bad rows:", np.unique(r)[:10], "n_badrows:", len(np.unique(r)))
        print("  bad cols:", np.unique(c)[:20], "n_badcols:", len(np.unique(c)))
        i = (r[0], c[0])
        print(f"  sample got={y[i]:.6f} ref={ref[i]:.6f} ratio={y[i]/(ref[i]+1e-30):.3f}")
        deg = np.diff(ch.column_offset.astype(np.int64))
        print("  deg of bad rows:", deg[np.unique(r)[:10]], "max deg overall:", deg.max())
