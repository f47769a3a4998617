def use_image():
    import imgmod  # provided by the built image's site dir
    return imgmod.IMG_MARKER
