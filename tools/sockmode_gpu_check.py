import multiprocessing, sys, time, tempfile
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))

def rank1(qname, filenames, result_q):
    try:
        import torch
        from ray_shuffling_data_loader_amd.dataset import ShufflingDataset
        ds = ShufflingDataset(filenames, 1, num_trainers=2, batch_size=20000,
                              rank=1, num_reducers=4, queue_name=qname)
        ds.set_epoch(0)
        total = rows_cuda = 0
        for b in ds:
            total += len(b)
            rows_cuda += int(b.device.type == "cuda")
        result_q.put(("ok", total, rows_cuda))
    except Exception as e:
        import traceback
        result_q.put(("err", str(e), traceback.format_exc()))

def main():
    import torch
    from ray_shuffling_data_loader_amd.data_generation import float_data_spec, generate_data
    from ray_shuffling_data_loader_amd.dataset import ShufflingDataset
    d = tempfile.mkdtemp()
    filenames, _ = generate_data(200000, 2, 1, 0.0, d, spec=float_data_spec(16), include_key=False)
    filenames = list(filenames)
    qname = f"gpu_sock_{time.time_ns()}"
    ctx = multiprocessing.get_context("spawn")
    rq = ctx.Queue()
    p = ctx.Process(target=rank1, args=(qname, filenames, rq))
    ds0 = ShufflingDataset(filenames, 1, num_trainers=2, batch_size=20000,
                           rank=0, num_reducers=4, queue_name=qname,
                           device=torch.device("cuda", 0))
    p.start()
    ds0.set_epoch(0)
    t0 = sum(len(b) for b in ds0)
    status = rq.get(timeout=180)
    p.join(timeout=60)
    print("rank0 rows:", t0, "rank1:", status)
    assert status[0] == "ok", status
    assert t0 + status[1] == 200000
    print("SOCKET MODE GPU: PASS")

if __name__ == "__main__":
    main()
