import sys, pathlib, time
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
import torch.nn.functional as F
import d9d_amd.module.block.attention.linear.gated_deltanet as gd

torch.manual_seed(0)
B, H, S, D = 8, 16, 4096, 64
q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).float()
k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).float()
v = (torch.randn(B, H, S, D, device="cuda") * 0.5).float()
beta = torch.rand(B, H, S, device="cuda")
g = -torch.rand(B, H, S, device="cuda") * 0.2
do = torch.randn(B, H, S, D, device="cuda")


def timed(fn, n=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


print("total: %.2f ms" % timed(lambda: gd._chunk_gdn_backward(q, k, v, beta, g, do)))


def pieces():
    C = 64
    Dk = Dv = D
    nc = S // C
    Qc = q.view(B, H, nc, C, Dk); Kc = k.view(B, H, nc, C, Dk); Vc = v.view(B, H, nc, C, Dv)
    dOc = do.view(B, H, nc, C, Dv); bc = beta.view(B, H, nc, C)
    gc = g.view(B, H, nc, C).cumsum(-1); E = gc.exp(); g_tot = gc[..., -1:]; Eend = g_tot.exp()
    w = (g_tot - gc).exp()
    ratio = torch.exp(gc.unsqueeze(-1) - gc.unsqueeze(-2)).tril(0)
    kk = torch.matmul(Kc, Kc.transpose(-1, -2))
    M = (bc.unsqueeze(-1) * kk * ratio).tril(-1)
    eye = torch.eye(C, device=q.device)
    A = M + eye
    qk = torch.matmul(Qc, Kc.transpose(-1, -2)); N = (qk * ratio).tril(0)
    EK = E.unsqueeze(-1) * Kc; EQ = E.unsqueeze(-1) * Qc; wK = w.unsqueeze(-1) * Kc
    return dict(Qc=Qc, Kc=Kc, Vc=Vc, dOc=dOc, bc=bc, gc=gc, E=E, Eend=Eend, w=w,
                ratio=ratio, kk=kk, M=M, A=A, qk=qk, N=N, EK=EK, EQ=EQ, wK=wK,
                nc=nc, C=C, Dk=Dk, Dv=Dv)


print("prologue: %.2f ms" % timed(pieces))
P = pieces()


def fwd_scan():
    nc, C, Dk, Dv = P["nc"], P["C"], P["Dk"], P["Dv"]
    S0s = torch.empty(B, H, nc, Dk, Dv, device=q.device)
    Rs = torch.empty(B, H, nc, C, Dv, device=q.device)
    state = torch.zeros(B, H, Dk, Dv, device=q.device)
    for i in range(nc):
        S0s[:, :, i] = state
        rhs = P["bc"][:, :, i].unsqueeze(-1) * (P["Vc"][:, :, i] - torch.matmul(P["EK"][:, :, i], state))
        R = torch.linalg.solve_triangular(P["A"][:, :, i], rhs, upper=False)
        Rs[:, :, i] = R
        state = P["Eend"][:, :, i].unsqueeze(-1) * state + torch.matmul(P["wK"][:, :, i].transpose(-1, -2), R)
    return S0s, Rs


print("fwd scan: %.2f ms" % timed(fwd_scan))
S0s, Rs = fwd_scan()
AT = P["A"].transpose(-1, -2)


def rev_scan():
    nc = P["nc"]
    dRs = torch.empty_like(Rs); drhss = torch.empty_like(Rs); dS0s = torch.empty_like(S0s)
    dS = torch.zeros(B, H, P["Dk"], P["Dv"], device=q.device)
    for i in range(nc - 1, -1, -1):
        dR = torch.matmul(P["wK"][:, :, i], dS) + torch.matmul(P["N"][:, :, i].transpose(-1, -2), P["dOc"][:, :, i])
        drhs = torch.linalg.solve_triangular(AT[:, :, i], dR, upper=True)
        dT = -(P["bc"][:, :, i].unsqueeze(-1) * drhs)
        dS = (P["Eend"][:, :, i].unsqueeze(-1) * dS
              + torch.matmul(P["EK"][:, :, i].transpose(-1, -2), dT)
              + torch.matmul(P["EQ"][:, :, i].transpose(-1, -2), P["dOc"][:, :, i]))
        dRs[:, :, i] = dR; drhss[:, :, i] = drhs; dS0s[:, :, i] = dS
    return dRs, drhss, dS0s


print("rev scan: %.2f ms" % timed(rev_scan))
