import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, sys
import zaremba_amd._hip as ext

mode = sys.argv[1] if len(sys.argv) > 1 else "nograph"
dev = torch.device("cuda:0")
T, B, H = 6, 20, 1500
KS = (H + 31) // 32
torch.manual_seed(0)
gx = (torch.randn(T, B, 4*H, device=dev)*0.5).to(torch.bfloat16)
W_h = (torch.randn(4*H, H, device=dev)*0.02).to(torch.bfloat16)
WhP = torch.empty(((H+15)//16)*4*KS*64*8, device=dev, dtype=torch.bfloat16)
ext.pack_gated_w(W_h, WhP, H, 4, H)
h_all = torch.zeros(T+1, B, H, device=dev, dtype=torch.bfloat16)
h_pack = torch.zeros(T+1, KS*2*64*8, device=dev, dtype=torch.bfloat16)
c_all = torch.zeros(T+1, B, H, device=dev, dtype=torch.float32)
gates = torch.empty(T, B, 4*H, device=dev, dtype=torch.bfloat16)
hgran = torch.zeros(32, device=dev, dtype=torch.int64)
abort = torch.zeros(1, device=dev, dtype=torch.int32)
h_all[0] = (torch.randn(B, H, device=dev)*0.3).to(torch.bfloat16)
c_all[0] = torch.randn(B, H, device=dev)*0.3
if mode == "nograph":
    ext.set_use_graphs(False)
print("launching", mode, flush=True)
ext.lstm_seq_fwd(gx, W_h, WhP, h_all, h_pack, c_all, gates, hgran, abort)
torch.cuda.synchronize()
print("done, abort =", abort.item(), flush=True)
print("h_all[1] sample:", h_all[1,0,:4].float().tolist(), flush=True)
# second call (replay path)
ext.lstm_seq_fwd(gx, W_h, WhP, h_all, h_pack, c_all, gates, hgran, abort)
torch.cuda.synchronize()
print("second call done, abort =", abort.item(), flush=True)
