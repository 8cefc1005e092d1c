import sys; sys.path.insert(0, '/root/repo')
import torch
from deepdfa_amd.models.t5 import T5Config, T5ForConditionalGeneration
import deepdfa_amd.models.t5 as t5mod

torch.manual_seed(0)
cfg = T5Config(num_layers=3, num_decoder_layers=3, d_model=128, d_ff=256,
               num_heads=2, vocab_size=500, dropout_rate=0.0)
m = T5ForConditionalGeneration(cfg).to("cuda:0")
m.train()
# spy on the sink + attention path
orig_apply = t5mod._BiasGradSink.apply
def spy(x, bias, accum):
    print("SINK engaged; bias req_grad:", bias.requires_grad, "accum id", id(accum))
    return orig_apply(x, bias, accum)
t5mod._BiasGradSink.apply = spy
from deepdfa_amd.ops import transformer as tr
orig_qkv = tr.flash_attention_qkv
def spy2(qkv, H, valid=None, bias=None, scale=1.0, causal=False, dropout_p=0.0, bias_accum=None):
    print("flash qkv: bias", None if bias is None else bias.shape, "accum?", bias_accum is not None)
    return orig_qkv(qkv, H, valid=valid, bias=bias, scale=scale, causal=causal,
                    dropout_p=dropout_p, bias_accum=bias_accum)
tr.flash_attention_qkv = spy2
t5mod.flash_attention_qkv = spy2 if hasattr(t5mod, 'flash_attention_qkv') else None
ids = torch.randint(3, cfg.vocab_size, (2, 64), device="cuda:0")
dec = m(ids, labels=ids, output_hidden_only=True)
dec.float().pow(2).mean().backward()
w = m.encoder.block[0].layer[0].SelfAttention.relative_attention_bias.weight
print("enc bias grad sum:", float(w.grad.abs().sum()) if w.grad is not None else None)
