import torch  # load torch libs before the extensions
