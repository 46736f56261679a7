"""Host wrapper for the selected-units sampling kernel (K7): marshals the
head's weights (bf16 rows) and per-row state, draws the per-step uniforms
with torch's RNG (reproducible under torch.manual_seed), and launches ONE
kernel for the whole data-dependent loop."""
import torch

from . import hip_ext


def su_sample(head, ae_base, key, avail_mask, su_mask, entity_num,
              temperature, uniforms=None):
    """Returns (logits (B,64,N1) fp32, results (B,64) long, num (B,) long).

    ``head`` is a models.alphastar.heads.SelectedUnitsHead (weights source).
    """
    ext = hip_ext.maybe_ext(ae_base)
    B, N1 = key.shape[0], key.shape[1]
    if uniforms is None:
        uniforms = torch.rand(B, 64, device=ae_base.device)
    cell = head.lstm.layers[0].cell
    args = dict(
        ae_base=ae_base.contiguous().float(),
        keys=key.detach().bfloat16().contiguous(),
        avail=avail_mask.to(torch.uint8).contiguous(),
        su_mask=su_mask.to(torch.uint8).contiguous(),
        entity_num=entity_num.to(torch.int32).contiguous(),
        uniforms=uniforms.float().contiguous(),
        Wq1=head.query_fc1[0].weight.detach().bfloat16().contiguous(),
        bq1=head.query_fc1[0].bias.detach().float(),
        Wq2=head.query_fc2[0].weight.detach().bfloat16().contiguous(),
        bq2=head.query_fc2[0].bias.detach().float(),
        Wih=cell.weight_ih.detach().bfloat16().contiguous(),
        Whh=cell.weight_hh.detach().bfloat16().contiguous(),
        lni_w=cell.layernorm_i.weight.detach().float(),
        lni_b=cell.layernorm_i.bias.detach().float(),
        lnh_w=cell.layernorm_h.weight.detach().float(),
        lnh_b=cell.layernorm_h.bias.detach().float(),
        lnc_w=cell.layernorm_c.weight.detach().float(),
        lnc_b=cell.layernorm_c.bias.detach().float(),
        We1=head.embed_fc1[0].weight.detach().bfloat16().contiguous(),
        be1=head.embed_fc1[0].bias.detach().float(),
        We2=head.embed_fc2[0].weight.detach().bfloat16().contiguous(),
        be2=head.embed_fc2[0].bias.detach().float(),
    )
    logits, results, num = ext.su_sample(*args.values(), float(temperature))
    return logits[:, :, :N1], results.long(), num.long()
