% Build the SPLATT MEX bindings against bin/libsplatt.so.
% Works in Octave (mkoctfile --mex) and MATLAB (mex).
%   >> cd matlab; make_splatt
root = fileparts(fileparts(mfilename('fullpath')));
inc = ['-I' fullfile(root, 'csrc', 'capi')];
lnk = {['-L' fullfile(root, 'bin')], '-lsplatt', ...
       ['-Wl,-rpath,' fullfile(root, 'bin')]};
for src = {'splatt_load.c', 'splatt_free.c', 'splatt_cpd.c', ...
           'splatt_mttkrp.c'}
  fprintf('mex %s\n', src{1});
  mex(inc, lnk{:}, src{1});
end
