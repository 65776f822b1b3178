"""Out-of-core hash join (grace-hash spill): both sides stream through
device partition -> pinned-host buckets -> per-bucket device join."""
import numpy as np
import pytest

import oracle

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu


def test_external_hash_join_matches_inner():
    from spark_amd.external_join import external_hash_join
    bn, pn = 4_000_000, 6_000_000
    bkeys = oracle.gen_i64(seed=801, n=bn, range_=3_000_000)
    bpay = oracle.gen_i64(seed=802, n=bn)
    pkeys = oracle.gen_i64(seed=803, n=pn, range_=3_000_000)
    ppay = oracle.gen_i64(seed=804, n=pn)

    def bb():
        yield torch.from_numpy(bkeys).cuda(), {"bp": torch.from_numpy(bpay).cuda()}

    def pb():
        half = pn // 2
        yield (torch.from_numpy(pkeys[:half]).cuda(),
               {"pp": torch.from_numpy(ppay[:half]).cuda()})
        yield (torch.from_numpy(pkeys[half:]).cuda(),
               {"pp": torch.from_numpy(ppay[half:]).cuda()})

    got_k, got_bp, got_pp = [], [], []
    for k, bp, pp in external_hash_join(bb(), pb(), budget_rows=1_000_000,
                                        nbuckets=8):
        got_k.append(k.cpu().numpy())
        got_bp.append(bp["bp"].cpu().numpy())
        got_pp.append(pp["pp"].cpu().numpy())
    gk = np.concatenate(got_k)
    gbp = np.concatenate(got_bp)
    gpp = np.concatenate(got_pp)
    op, ob = oracle.join_inner(bkeys, pkeys)
    assert len(gk) == len(op)
    g = np.lexsort((gpp, gbp, gk))
    o = np.lexsort((ppay[op], bpay[ob], bkeys[ob]))
    assert (gk[g] == bkeys[ob][o]).all()
    assert (gbp[g] == bpay[ob][o]).all()
    assert (gpp[g] == ppay[op][o]).all()
