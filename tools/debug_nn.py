"""Micro-diagnostics for k_gemm_nn_splitk: K=32 single k-step, identity-ish A.

With A[i,k] = (k == i % 32), C[i,:] should equal B[i % 32, :].  For each
output row we find WHICH B row it actually matches -> reveals the index
routing of the staging / fragment / epilogue paths.
"""
import torch

from code2vec_amd.ops import hip_ext

ext = hip_ext(required=True)


def main():
    N, M, K = 256, 384, 32
    A = torch.zeros(N, K, device='cuda')
    for i in range(N):
        A[i, i % 32] = 1.0
    A = A.to(torch.bfloat16)
    # B: unique value per (k, c): k*1000 + c
    B = (torch.arange(K, device='cuda')[:, None] * 1000.0 +
         torch.arange(M, device='cuda')[None, :]).to(torch.bfloat16)
    C = ext.gemm_nn_splitk(A, B)
    Bf = B.float()
    ref = A.float() @ Bf

    print('C[0,0:6]   =', C[0, :6].tolist())
    print('ref[0,0:6] =', ref[0, :6].tolist())
    print('C[1,0:6]   =', C[1, :6].tolist())
    print('ref[1,0:6] =', ref[1, :6].tolist())
    print('C[0,16:22] =', C[0, 16:22].tolist())
    print('ref[0,16:22]=', ref[0, 16:22].tolist())

    # row routing: which B row does each C row equal?
    d = (C[:, None, :] - Bf[None, :, :]).abs().mean(-1)   # (N, K)
    near = d.min(1)
    print('rows 0..15 matched B row:', near.indices[:16].tolist())
    print('rows 0..15 match err:', [round(x, 2) for x in near.values[:16].tolist()])
    print('rows 16..31 matched B row:', near.indices[16:32].tolist())

    # column routing within a row: C[0, c] should be B[0, c] = c
    got = C[0, :32].tolist()
    print('C[0, c]-c for c in 0..31:', [round(g - c, 1) for c, g in enumerate(got)])

    # full-random small check with the same single-kstep shape
    torch.manual_seed(0)
    Ar = (torch.randn(N, K, device='cuda') * 0.1).to(torch.bfloat16)
    Br = (torch.randn(K, M, device='cuda') * 0.1).to(torch.bfloat16)
    Cr = ext.gemm_nn_splitk(Ar, Br)
    rr = Ar.float() @ Br.float()
    rel = (Cr - rr).abs().max().item() / rr.abs().max().item()
    print('random single-kstep rel err: %.3g' % rel)


if __name__ == '__main__':
    main()
