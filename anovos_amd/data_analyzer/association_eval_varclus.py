"""Variable clustering (VarClusHi) — parity with reference
data_analyzer/association_eval_varclus.py (450 LoC).

The reference's only at-scale computation is ONE covariance pass over
standardized features (association_eval_varclus.py:71-84, a Spark
RowMatrix.computeCovariance); everything after is driver-side numpy on
the KxK correlation matrix. Here the covariance comes from the fused K8
Gram kernel (ops/corr.py) and the class operates directly on that
matrix. The quartimax rotation (the reference imports factor_analyzer's
Rotator) is implemented as the standard orthomax(gamma=0) SVD iteration.

Algorithm (reference :267-383): start with one cluster; repeatedly split
the cluster with the largest 2nd eigenvalue via its top-2 PCs +
quartimax rotation, then NCS/search reassignment; stop when every 2nd
eigenvalue <= maxeigval2 (or maxclus reached).
"""

from __future__ import annotations

import collections
import math
import random
from typing import List, Optional

import numpy as np
import pandas as pd

ClusInfo = collections.namedtuple("ClusInfo", ["clus", "eigval1", "eigval2", "eigvecs", "varprop"])


def quartimax_rotation(L: np.ndarray, max_iter: int = 500, tol: float = 1e-9) -> np.ndarray:
    """Orthomax rotation with gamma=0 (quartimax) via the classic SVD
    iteration (equivalent to factor_analyzer Rotator(method='quartimax'))."""
    n, k = L.shape
    R = np.eye(k)
    d = 0.0
    for _ in range(max_iter):
        Lr = L @ R
        u, s, vt = np.linalg.svd(L.T @ (Lr**3))
        R = u @ vt
        d_new = float(np.sum(s))
        if d_new <= d * (1 + tol):
            break
        d = d_new
    return L @ R


class VarClusHi:
    """Driver-side variable clustering over a precomputed correlation
    (covariance-of-standardized) matrix."""

    def __init__(self, corr_df: pd.DataFrame, feat_list: Optional[List[str]] = None, maxeigval2: float = 1, maxclus: Optional[int] = None, n_rs: int = 0):
        self.feat_list = feat_list or list(corr_df.columns)
        self.corr_df = corr_df.loc[self.feat_list, self.feat_list]
        self.maxeigval2 = maxeigval2
        self.maxclus = maxclus
        self.n_rs = n_rs

    def correig(self, feat_list: List[str], n_pcs: int = 2):
        """Top n_pcs eig of the correlation submatrix (reference :90-144)."""
        if len(feat_list) <= 1:
            corr = [len(feat_list)]
            eigvals = [len(feat_list)] + [0] * (n_pcs - 1)
            eigvecs = np.array([[len(feat_list)]])
            varprops = [sum(eigvals)]
            corr_df = pd.DataFrame(corr, columns=feat_list, index=feat_list) if feat_list else pd.DataFrame()
            return eigvals, eigvecs, corr_df, varprops
        corr = self.corr_df.loc[feat_list, feat_list].values
        raw_eigvals, raw_eigvecs = np.linalg.eigh(corr)
        idx = np.argsort(raw_eigvals)[::-1]
        eigvals, eigvecs = raw_eigvals[idx], raw_eigvecs[:, idx]
        eigvals, eigvecs = eigvals[:n_pcs], eigvecs[:, :n_pcs]
        varprops = eigvals / sum(raw_eigvals)
        return eigvals, eigvecs, pd.DataFrame(corr, columns=feat_list, index=feat_list), varprops

    def _calc_tot_var(self, *clusters):
        tot_len = tot_var = tot_prop = 0
        for clus in clusters:
            if clus == []:
                continue
            c_len = len(clus)
            c_eigvals, _, _, c_varprops = self.correig(clus)
            tot_var += c_eigvals[0]
            tot_prop = (tot_prop * tot_len + c_varprops[0] * c_len) / (tot_len + c_len)
            tot_len += c_len
        return tot_var, tot_prop

    def _reassign(self, clus1, clus2, feat_list=None):
        if feat_list is None:
            feat_list = clus1 + clus2
        init_var = self._calc_tot_var(clus1, clus2)[0]
        fin_clus1, fin_clus2 = clus1[:], clus2[:]
        check_var = max_var = init_var
        while True:
            for feat in feat_list:
                new_clus1, new_clus2 = fin_clus1[:], fin_clus2[:]
                if feat in new_clus1:
                    new_clus1.remove(feat)
                    new_clus2.append(feat)
                elif feat in new_clus2:
                    new_clus1.append(feat)
                    new_clus2.remove(feat)
                else:
                    continue
                new_var = self._calc_tot_var(new_clus1, new_clus2)[0]
                if new_var > check_var:
                    check_var = new_var
                    fin_clus1, fin_clus2 = new_clus1[:], new_clus2[:]
            if max_var == check_var:
                break
            max_var = check_var
        return fin_clus1, fin_clus2, max_var

    def _reassign_rs(self, clus1, clus2, n_rs=0):
        feat_list = clus1 + clus2
        fin_clus1, fin_clus2, max_var = self._reassign(clus1, clus2)
        for _ in range(n_rs):
            random.shuffle(feat_list)
            rs1, rs2, rs_var = self._reassign(clus1, clus2, feat_list)
            if rs_var > max_var:
                max_var = rs_var
                fin_clus1, fin_clus2 = rs1, rs2
        return fin_clus1, fin_clus2, max_var

    def varclus(self):
        c_eigvals, c_eigvecs, c_corrs, c_varprops = self.correig(self.feat_list)
        self.corrs = c_corrs
        self.clusters = collections.OrderedDict(
            [(0, ClusInfo(clus=self.feat_list, eigval1=c_eigvals[0], eigval2=c_eigvals[1], eigvecs=c_eigvecs, varprop=c_varprops[0]))]
        )
        while True:
            if self.maxclus is not None and len(self.clusters) >= self.maxclus:
                break
            idx = max(self.clusters, key=lambda x: self.clusters.get(x).eigval2)
            if self.clusters[idx].eigval2 > self.maxeigval2:
                split_clus = self.clusters[idx].clus
                c_eigvals, c_eigvecs, split_corrs, _ = self.correig(split_clus)
            else:
                break
            if c_eigvals[1] > self.maxeigval2:
                clus1, clus2 = [], []
                r_eigvecs = quartimax_rotation(np.asarray(c_eigvecs))
                comb_sigmas = np.sqrt(np.diag(r_eigvecs.T @ split_corrs.values @ r_eigvecs))
                for feat in split_clus:
                    comb_cov1 = np.dot(r_eigvecs[:, 0], split_corrs[feat].values.T)
                    comb_cov2 = np.dot(r_eigvecs[:, 1], split_corrs[feat].values.T)
                    corr_pc1 = comb_cov1 / comb_sigmas[0]
                    corr_pc2 = comb_cov2 / comb_sigmas[1]
                    (clus1 if abs(corr_pc1) > abs(corr_pc2) else clus2).append(feat)
                fin_clus1, fin_clus2, _ = self._reassign_rs(clus1, clus2, self.n_rs)
                c1 = self.correig(fin_clus1)
                c2 = self.correig(fin_clus2)
                self.clusters[idx] = ClusInfo(clus=fin_clus1, eigval1=c1[0][0], eigval2=c1[0][1], eigvecs=c1[1], varprop=c1[3][0])
                self.clusters[len(self.clusters)] = ClusInfo(clus=fin_clus2, eigval1=c2[0][0], eigval2=c2[0][1], eigvecs=c2[1], varprop=c2[3][0])
            else:
                break
        return self

    def rsquare(self) -> pd.DataFrame:
        """[Cluster, Variable, RS_Own, RS_NC, RS_Ratio] — reference :384-450."""
        rows = []
        sigmas = []
        for _, clusinfo in self.clusters.items():
            c_eigvec = clusinfo.eigvecs[:, 0]
            c_sigma = math.sqrt(np.dot(np.dot(c_eigvec, self.corr_df.loc[clusinfo.clus, clusinfo.clus].values), c_eigvec.T))
            sigmas.append(c_sigma)
        for i, clus_own in self.clusters.items():
            for feat in clus_own.clus:
                cov_own = np.dot(clus_own.eigvecs[:, 0], self.corr_df.loc[feat, clus_own.clus].values.T)
                if len(clus_own.clus) == 1 and feat == clus_own.clus[0]:
                    rs_own = 1
                else:
                    rs_own = (cov_own / sigmas[i]) ** 2
                rs_others = []
                for j, clus_other in self.clusters.items():
                    if j == i:
                        continue
                    cov_other = np.dot(clus_other.eigvecs[:, 0], self.corr_df.loc[feat, clus_other.clus].values.T)
                    rs_others.append((cov_other / sigmas[j]) ** 2)
                rs_nc = max(rs_others) if rs_others else 0
                rows.append([i, feat, rs_own, rs_nc, (1 - rs_own) / (1 - rs_nc) if rs_nc != 1 else float("inf")])
        return pd.DataFrame(rows, columns=["Cluster", "Variable", "RS_Own", "RS_NC", "RS_Ratio"])
