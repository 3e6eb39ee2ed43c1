import torch

from libai_amd.evaluation import BleuEvaluator, PPLEvaluator, RegEvaluator, corpus_bleu


def test_corpus_bleu_perfect_and_disjoint():
    assert corpus_bleu([[1, 2, 3, 4, 5]], [[1, 2, 3, 4, 5]]) == 1.0
    assert corpus_bleu([[9, 9, 9, 9, 9]], [[1, 2, 3, 4, 5]]) == 0.0


def test_bleu_evaluator_protocol():
    ev = BleuEvaluator()
    ev.reset()
    ev.process({"labels": torch.tensor([[1, 2, 3, 4]])},
               {"sequences": torch.tensor([[1, 2, 3, 4]])})
    res = ev.evaluate()
    assert res["bleu"]["bleu"] == 100.0


def test_reg_evaluator():
    ev = RegEvaluator()
    ev.reset()
    x = torch.randn(64)
    ev.process({"labels": x}, {"prediction_scores": x})
    res = ev.evaluate()
    assert res["regression"]["mse"] < 1e-10
    assert abs(res["regression"]["pearson"] - 1.0) < 1e-4


def test_ppl_evaluator():
    ev = PPLEvaluator()
    ev.reset()
    ev.process({"labels": torch.zeros(2, 8)}, {"lm_loss": torch.tensor(1.0)})
    res = ev.evaluate()
    assert abs(res["ppl"]["nll"] - 1.0) < 1e-6
