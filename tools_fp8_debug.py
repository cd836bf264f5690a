import sys, torch
sys.path.insert(0, '.')
from vizier_amd._src.ops import dispatch as ops
ext = ops.require_ext()
x = torch.rand(64, 16).cuda()
ls = torch.full((16,), 0.5).cuda()
K = ext.gram_matern52_fp8(x, x, ls, 2.0)
diag = torch.diagonal(K)
print('diag err max:', float((diag - 4.0).abs().max()))
print('diag sample:', diag[:6].tolist())
from vizier_amd._src.gp.matern import gram_matern52
want = gram_matern52(x.cpu().double(), None, ls.cpu().double(), torch.tensor(2.0).double())
print('cross err max:', float((K.cpu().double() - want).abs().max()))
print('cross err transposed:', float((K.cpu().double() - want.T).abs().max()))
