import os


def read_mounted():
    path = "/vol/data/models/weights.txt"
    exists = os.path.exists(path)
    content = open(path).read() if exists else ""
    return {"content": content, "exists": exists}


def write_mounted():
    with open("/vol/data/from-container.txt", "w") as f:
        f.write("persisted")
    return True


def gpu_in_native():
    import os

    import torch

    ok = torch.cuda.is_available()
    devs = os.listdir("/dev")
    x = None
    if ok:
        x = float((torch.ones(64, device="cuda") * 2).sum().item())
    return {
        "cuda": ok,
        "kfd": "kfd" in devs,
        "dri": os.path.isdir("/dev/dri"),
        "sum": x,
        "vol": os.path.exists("/vol/data/models/weights.txt"),
    }
