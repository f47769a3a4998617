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
