def double(x):
    return x * 2
