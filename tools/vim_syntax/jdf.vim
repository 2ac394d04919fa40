" Vim syntax for PaRSEC-style .jdf files (parsec_amd.ptg dialect).
" Counterpart of the reference's tools/vim_syntax. Install:
"   mkdir -p ~/.vim/syntax && cp jdf.vim ~/.vim/syntax/
"   echo 'au BufRead,BufNewFile *.jdf set filetype=jdf' >> ~/.vim/ftdetect/jdf.vim
if exists("b:current_syntax")
  finish
endif

syn include @jdfC syntax/c.vim

" extern "C" %{ ... %} prologue/epilogue and BODY blocks are C
syn region jdfProlog matchgroup=jdfDelim start="%{" end="%}" contains=@jdfC
syn region jdfBody matchgroup=jdfKeyword start="^BODY" end="^END" contains=@jdfC,jdfBodyProps
syn region jdfBodyProps contained start="\[" end="\]" contains=jdfProp

syn keyword jdfKeyword CTL READ WRITE RW NEW NULL
syn match jdfArrow "<-\|->"
syn match jdfRange "\.\."
syn match jdfPartition "^\s*:\s*\w\+"
syn region jdfProps start="\[" end="\]" contains=jdfProp oneline
syn match jdfProp contained "\w\+\s*="
syn match jdfTaskDef "^\w\+\s*(\s*[a-zA-Z_, ]*)\s*$"
syn match jdfComment "//.*$"
syn region jdfComment start="/\*" end="\*/"
syn region jdfString start=+"+ skip=+\\"+ end=+"+

hi def link jdfKeyword Keyword
hi def link jdfArrow Operator
hi def link jdfRange Operator
hi def link jdfPartition Special
hi def link jdfProp Identifier
hi def link jdfTaskDef Function
hi def link jdfComment Comment
hi def link jdfString String
hi def link jdfDelim PreProc

let b:current_syntax = "jdf"
