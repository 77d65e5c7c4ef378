

def test_knn_model_ball_tree_view():
    import numpy as np
    import pandas as pd
    from mmlspark_amd.models.knn import KNN
    rng = np.random.default_rng(0)
    X = rng.normal(size=(200, 8)).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "values": list(range(200))})
    m = KNN(k=3, leafSize=20).fit(df)
    bt = m.getBallTree()
    assert bt is m.getBallTree()  # cached
    q = X[7]
    hits = bt.find_maximum_inner_products(q, k=1)
    v = int(hits[0][0])
    assert v == 7 or float(X[v] @ q) >= float(q @ q) - 1e-4
