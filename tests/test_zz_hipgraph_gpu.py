"""hipGraph serving capture test — kept in a file that sorts LAST:
an aborted stream capture can poison the CUDA context, and this test
must not take the rest of the GPU suite down with it."""

import numpy as np
import pytest
import torch


@pytest.mark.gpu
@pytest.mark.xfail(strict=False,
                   reason="hipGraph serving path pending first GPU "
                          "validation (round-2 block)")
def test_engine_hipgraph_matches_eager(shapes_config, monkeypatch):
    """hipGraph replay of the serving forward must equal eager output
    for the same input (engine._forward capture/replay path)."""
    from maskrcnn_amd.serving.engine import InferenceEngine
    torch.manual_seed(0)
    eager = InferenceEngine(shapes_config)
    monkeypatch.setenv("MASKRCNN_SERVING_HIPGRAPH", "1")
    graphed = InferenceEngine(shapes_config)
    graphed.model.load_state_dict(eager.model.state_dict())
    assert graphed.use_hip_graph
    H = shapes_config.IMAGE_SHAPE[0]
    metas = np.zeros((1, shapes_config.IMAGE_META_SIZE), np.float32)
    metas[:, 4:7] = (H, H, 3)
    metas[:, 7:11] = (0, 0, H, H)
    for trial in range(2):  # second input exercises replay, not capture
        images = (np.random.rand(1, H, H, 3).astype(np.float32) * 50 +
                  trial)
        o_e = eager.predict(images, metas)
        o_g = graphed.predict(images, metas)
        for k in o_e:
            np.testing.assert_allclose(o_g[k], o_e[k], atol=1e-3,
                                       rtol=1e-3)
    assert len(graphed._graphs) == 1
