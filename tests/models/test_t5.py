

def test_t5_gated_mlp_variants():
    """MT5-style gated MLP: gelu- and silu-gated both train."""
    import torch

    from libai_amd.models import T5ForPreTraining

    for act in ("gelu", "silu"):
        torch.manual_seed(0)
        m = T5ForPreTraining(vocab_size=128, hidden_size=64, hidden_layers=2,
                             num_attention_heads=4, intermediate_size=128,
                             max_position_embeddings=64, mlp_type="gated",
                             activation=act)
        out = m(encoder_input_ids=torch.randint(0, 128, (2, 16)),
                decoder_input_ids=torch.randint(0, 128, (2, 8)),
                lm_labels=torch.randint(0, 128, (2, 8)),
                loss_mask=torch.ones(2, 8, dtype=torch.long))
        out["masked_lm_loss"].backward()
        assert torch.isfinite(out["masked_lm_loss"])
        gate_w = m.t5_model.encoder_layers[0].mlp.gate_up_proj.weight
        assert gate_w.grad is not None and torch.isfinite(gate_w.grad).all()
