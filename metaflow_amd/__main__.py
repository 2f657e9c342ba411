from .cmd import main

main()
